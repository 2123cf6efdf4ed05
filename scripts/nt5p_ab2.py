import torch, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
tag = os.environ.get("GLOM_NT5P", "d")
for lbl, M, N, K, np_, epi in [("down 16384x512x2048", 16384, 512, 2048, 6, 0),
                               ("down small-M 2048x512x2048", 2048, 512, 2048, 6, 0),
                               ("up", 16384, 2048, 512, 6, 0)]:
    ms = ext.bench_gemm(M, N, K, 0, np_, epi, 30)
    tf = 2.0 * M * N * K * np_ / (ms * 1e-3) / 1e12
    print(f"NT5P={tag} {lbl}: {ms:.3f} ms {tf:.0f} TF")
