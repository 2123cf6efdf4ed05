import sys, os
sys.path.insert(0, os.getcwd())
import torch
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
torch.manual_seed(0)
dev, bf = 'cuda', torch.bfloat16
for B in (2, 8):
    N, L, d = 256, 6, 512
    m4 = 4 * d
    tokens = torch.randn(B, N, d, device=dev, dtype=bf)
    levels = torch.randn(B, N, L, d, device=dev, dtype=bf)
    w1 = torch.randn(L*m4, d, device=dev, dtype=bf)*0.02
    b1 = torch.randn(L*m4, device=dev, dtype=bf)*0.1
    w2 = torch.randn(L*d, m4, device=dev, dtype=bf)*0.02
    b2 = torch.randn(L*d, device=dev, dtype=bf)*0.1
    for mode in (0, 1):
        ext.set_gelu_pair(bool(mode))
        out = ext.grouped_ff_fwd(tokens, levels, None, w1, b1, w2, b2, 0)
        torch.cuda.synchronize()
        print(f"B={B} pair={mode}: ok", out[0].float().norm().item())
        if mode == 0:
            ref = [t.clone() for t in out]
        else:
            for a, b_, nm in zip(ref, out, ("Y","Hpre","Hact","td")):
                if a.numel():
                    print(f"  {nm} bitwise={torch.equal(a,b_)}")
print("DONE")
