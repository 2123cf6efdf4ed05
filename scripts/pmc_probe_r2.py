import torch, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
# round-2 kernels: nt8p (down-proj K=2048 shape) and the fused consensus
# fwd with the double-buffered AV (nt4 EPI_SOFTMAX)
ext.set_nt8p(True)
ext.bench_gemm(16384, 512, 2048, 0, 6, 0, 3)      # nt8p down-proj
lv = (torch.randn(64, 256, 6, 512, device="cuda") * 0.5).to(torch.bfloat16)
for _ in range(3):
    ext.consensus_fwd(lv, False, None)            # nt4 fused softmax+AV
torch.cuda.synchronize()
print("pmc r2 probe done")
