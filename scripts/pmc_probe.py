import torch, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
# one call each of the two flagship shapes, tiny rep count
ext.bench_gemm(16384, 2048, 512, 0, 6, 2, 3)    # up-fwd pair
ext.bench_gemm(16384, 512, 2048, 0, 6, 0, 3)    # down-fwd
ext.bench_gemm(2048, 512, 16384, 2, 6, 0, 3)    # dW1 TN split-K
torch.cuda.synchronize()
print("pmc probe done")
