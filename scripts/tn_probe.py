import torch, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
for lbl, M, N, K, np_ in [("dW1 headline (2048,512,16384)x6", 2048, 512, 16384, 6),
                          ("dW2 headline (512,2048,16384)x6", 512, 2048, 16384, 6),
                          ("dW1 stretch (4096,1024,8192)x12", 4096, 1024, 8192, 12),
                          ("dW2 stretch (1024,4096,8192)x12", 1024, 4096, 8192, 12)]:
    ms = ext.bench_gemm(M, N, K, 2, np_, 0, 20)
    tf = 2.0 * M * N * K * np_ / (ms * 1e-3) / 1e12
    print(f"TN {lbl}: {ms:.3f} ms {tf:.0f} TF")
