import sys, os, time
sys.path.insert(0, os.getcwd())
import torch
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
torch.manual_seed(0)
dev, bf = 'cuda', torch.bfloat16
B, N, L, d = 64, 256, 6, 512
lv = (torch.randn(B, N, L, d, device=dev) * 0.5).to(bf)
out, probs, rnorm = ext.consensus_fwd(lv, False, None)
# fp32 reference
l32 = lv.float()
q = l32
k = torch.nn.functional.normalize(l32, dim=-1)
sim = torch.einsum("bild,bjld->blij", q, k) * d ** -0.5
eye = torch.eye(N, device=dev, dtype=torch.bool)
sim = sim.masked_fill(eye.view(1, 1, N, N), -5e-4)
attn = sim.softmax(dim=-1)
ref = torch.einsum("blij,bjld->bild", attn, l32)
err = ((out.float() - ref).norm() / ref.norm()).item()
print("consensus fwd relerr:", err)
assert err < 2e-2
# determinism screen
base = ext.consensus_fwd(lv, False, None)
for _ in range(5):
    cur = ext.consensus_fwd(lv, False, None)
    for a, b in zip(base, cur):
        assert torch.equal(a, b)
print("determinism OK")
def t(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3
ms = t(lambda: ext.consensus_fwd(lv, False, None))
fl = 2.0*B*L*N*N*d*2
print(f"consensus fwd: {ms:.3f} ms  {fl/(ms/1e3)/1e12:.1f} TF-equivalent")
