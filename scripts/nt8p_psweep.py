import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
torch.manual_seed(0)
SHAPES = [
    ("up-proj", 16384, 2048, 512, 0, 6, 0, 30),
    ("dH gelugrad", 16384, 2048, 512, 0, 5, 1, 30),
    ("down-proj", 16384, 512, 2048, 0, 6, 0, 30),
    ("square", 16384, 2048, 2048, 0, 6, 0, 15),
]
# GLOM_NT8P_P is read once (static) - sweep via separate invocation arg
P = os.environ.get("GLOM_NT8P_P", "auto")
for name, M, N, K, lay, G, epi, reps in SHAPES:
    fl = 2.0 * M * N * K * G
    ext.set_nt8p(True)
    r = sorted(fl / (ext.bench_gemm(M, N, K, lay, G, epi, reps) / 1e3) / 1e12
               for _ in range(3))
    print(f"P={P} {name:12s}: {r[1]:6.1f} TF (min {r[0]:.1f})")
