"""A/B: standalone k_gelu pass vs EPI_GELU_PAIR fused into the nt5p
register epilogue. Correctness gate: bitwise (gelu of the bf16-rounded
pre-activation both ways)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
torch.manual_seed(0)
dev, bf = 'cuda', torch.bfloat16
B, N, L, d = 64, 256, 6, 512
m4 = 4 * d
tokens = torch.randn(B, N, d, device=dev, dtype=bf)
levels = torch.randn(B, N, L, d, device=dev, dtype=bf)
w1 = torch.randn(L*m4, d, device=dev, dtype=bf)*0.02
b1 = torch.randn(L*m4, device=dev, dtype=bf)*0.1
w2 = torch.randn(L*d, m4, device=dev, dtype=bf)*0.02
b2 = torch.randn(L*d, device=dev, dtype=bf)*0.1

ext.set_gelu_pair(False)
ref = ext.grouped_ff_fwd(tokens, levels, None, w1, b1, w2, b2, 0)
ext.set_gelu_pair(True)
out = ext.grouped_ff_fwd(tokens, levels, None, w1, b1, w2, b2, 0)
for a, b, nm in zip(ref, out, ("Y", "Hpre", "Hact", "tdin")):
    if a.numel() == 0: continue
    print(f"{nm}: bitwise={torch.equal(a, b)}")
    assert torch.equal(a, b), nm

def t(fn, n=30):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3
for mode in (0, 1, 0, 1):
    ext.set_gelu_pair(bool(mode))
    ms = t(lambda: ext.grouped_ff_fwd(tokens, levels, None, w1, b1, w2, b2, 0))
    print(f"gelu_pair={mode}: ff fwd {ms:.3f} ms")
