"""Within-process interleaved A/B: nt8p (8-phase 256^2) vs incumbent
(nt5p / nt_fast) on the model NT shapes (guide methodology rules 24/25:
interleaved rounds in ONE process, random operands)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
torch.manual_seed(0)

# (M, N, K, layout(NT=0), G, epilogue, reps)
SHAPES = [
    ("up-proj fwd",  16384, 2048, 512, 0, 6, 0, 30),
    ("dH bwd gelugrad", 16384, 2048, 512, 0, 5, 1, 30),
    ("down-proj fwd", 16384, 512, 2048, 0, 6, 0, 30),
    ("square 2048",  16384, 2048, 2048, 0, 6, 0, 15),
    ("stretch up",   4096, 4096, 1024, 0, 12, 0, 15),
]
ROUNDS = 5
for name, M, N, K, lay, G, epi, reps in SHAPES:
    fl = 2.0 * M * N * K * G
    res = {0: [], 1: []}
    for r in range(ROUNDS):
        for mode in (0, 1):
            ext.set_nt8p(mode == 1)
            ms = ext.bench_gemm(M, N, K, lay, G, epi, reps)
            res[mode].append(fl / (ms / 1e3) / 1e12)
    a = sorted(res[0]); b = sorted(res[1])
    med = lambda x: x[len(x)//2]
    print(f"{name:18s} M{M} N{N} K{K} G{G} epi{epi}: "
          f"incumbent {med(a):6.1f} TF (min {a[0]:.1f})  "
          f"nt8p {med(b):6.1f} TF (min {b[0]:.1f})  "
          f"ratio {med(b)/med(a):.3f}")
