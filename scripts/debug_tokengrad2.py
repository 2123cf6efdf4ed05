import torch, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd import Glom
from glom_pytorch_amd.ops import _load_extension
from glom_pytorch_amd.ops.functional import GroupedFFFn
ext = _load_extension()
DEV = "cuda:0"
torch.manual_seed(0)
bf = torch.bfloat16

# Case 1: non-leaf tokens through a Linear, single bu Function
B, N, L, d = 2, 16, 3, 64; m4 = 4*d
lin = torch.nn.Linear(10, d).to(DEV, bf)
raw = torch.randn(B, N, 10, device=DEV, dtype=bf)
tokens = lin(raw)
levels = torch.randn(B, N, L, d, device=DEV, dtype=bf, requires_grad=True)
w1 = (torch.randn(L*m4, d, device=DEV, dtype=bf)*0.05).requires_grad_()
b1 = torch.randn(L*m4, device=DEV, dtype=bf, requires_grad=True)
w2 = (torch.randn(L*d, m4, device=DEV, dtype=bf)*0.05).requires_grad_()
b2 = torch.randn(L*d, device=DEV, dtype=bf, requires_grad=True)
out = GroupedFFFn.apply(tokens, levels, None, w1, b1, w2, b2, 0)
out.float().pow(2).mean().backward()
print("case1 lin.weight grad norm:", lin.weight.grad.float().norm().item())

# Case 2: direct ext backward call, check dTokens
tokens2 = torch.randn(B, N, d, device=DEV, dtype=bf)
Y, Hpre = ext.grouped_ff_fwd(tokens2, levels.detach(), None, w1.detach(), b1.detach(), w2.detach(), b2.detach(), 0)
dY = torch.randn_like(Y)
outs = ext.grouped_ff_bwd(dY, tokens2, levels.detach(), None, w1.detach(), w2.detach(), Hpre, 0)
print("case2 dTokens norm:", outs[0].float().norm().item(),
      "dLevels norm:", outs[1].float().norm().item(),
      "dW1 norm:", outs[2].float().norm().item())

# Case 3: model context — patch the Function to print
orig_bwd = GroupedFFFn.backward
class Dbg(GroupedFFFn):
    pass
def bwd(ctx, dY):
    r = orig_bwd(ctx, dY)
    if ctx.mode == 0:
        print("  full-graph bu bwd: dY", dY.float().norm().item(),
              "dTokens", r[0].float().norm().item() if r[0] is not None else None,
              "needs", ctx.needs_input_grad[:3])
    return r
GroupedFFFn.backward = staticmethod(bwd)
m = Glom(dim=64, levels=3, image_size=32, patch_size=8).to(DEV, bf)
img = torch.randn(2, 3, 32, 32, device=DEV, dtype=bf)
o = m(img, iters=2, return_all=True)
loss = o[2, :, :, -1].float().pow(2).mean()
loss.backward()
print("case3 image_to_tokens w grad:", m.image_to_tokens[1].weight.grad.float().norm().item())
