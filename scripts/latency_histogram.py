"""Serving latency distribution: N sequential B=1 forwards under hipGraph
replay; reports p50/p90/p99/max (per-call CUDA-event timing) plus a
sustained-load check (first vs last decile of a long run — catches DVFS
sag or allocator drift)."""
import sys, os, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from glom_pytorch_amd import Glom

N = int(sys.argv[1]) if len(sys.argv) > 1 else 3000
torch.manual_seed(0)
m = Glom(dim=512, levels=6, image_size=224, patch_size=14).to(
    "cuda", torch.bfloat16)
m.enable_graphs()
x = torch.randn(1, 3, 224, 224, device="cuda", dtype=torch.bfloat16)
with torch.no_grad():
    for _ in range(20):
        m(x, iters=12)
    torch.cuda.synchronize()
    times = []
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    for _ in range(N):
        s.record()
        m(x, iters=12)
        e.record()
        e.synchronize()
        times.append(s.elapsed_time(e))
arrival = times[:]
times.sort()
q = lambda p: times[min(N - 1, int(p * N))]
d = N // 10
first, last = sum(arrival[:d]) / d, sum(arrival[-d:]) / d
print(json.dumps({
    "n": N, "p50_ms": round(q(0.5), 3), "p90_ms": round(q(0.9), 3),
    "p99_ms": round(q(0.99), 3), "max_ms": round(times[-1], 3),
    "first_decile_mean_ms": round(first, 3),
    "last_decile_mean_ms": round(last, 3),
}))
