import sys, os; sys.path.insert(0, os.getcwd())
import torch
from glom_pytorch_amd import Glom

DEV = "cuda:0"
torch.manual_seed(0)
cfg = dict(dim=512, levels=6, image_size=224, patch_size=14)
m32 = Glom(**cfg).to(DEV); m32.force_eager = True
mbf = Glom(**cfg).to(DEV)
mbf.load_state_dict(m32.state_dict())
mbf = mbf.to(torch.bfloat16)

img = torch.randn(2, 3, 224, 224, device=DEV)

# hook the tokens grad on the native side
from glom_pytorch_amd.ops import functional as Fn
orig_bwd = Fn.PatchEmbedFn.backward
def dbg_bwd(ctx, dTokens):
    print("dTokens norm:", dTokens.float().norm().item(),
          "shape", tuple(dTokens.shape), "dtype", dTokens.dtype,
          "contig", dTokens.is_contiguous())
    out = orig_bwd(ctx, dTokens)
    print("dW norm:", out[1].float().norm().item(),
          "dB norm:", out[2].float().norm().item())
    X, w = ctx.saved_tensors
    # reference dW from the same dTokens
    ref = dTokens.reshape(-1, dTokens.shape[-1]).float().t() @ X.reshape(-1, X.shape[-1]).float()
    print("ref dW norm:", ref[:, :588].norm().item(),
          "relerr:", ((out[1].float() - ref[:, :588]).norm() / ref[:, :588].norm().clamp_min(1e-12)).item())
    return out
Fn.PatchEmbedFn.backward = dbg_bwd

for iters, ra in [(1, False), (3, True)]:
    for p in list(m32.parameters()) + list(mbf.parameters()):
        p.grad = None
    print(f"== iters={iters} return_all={ra}")
    if ra:
        ref = m32(img, iters=iters, return_all=True)
        out = mbf(img.to(torch.bfloat16), iters=iters, return_all=True)
        ref[2, :, :, -1].float().pow(2).mean().backward()
        out[2, :, :, -1].float().pow(2).mean().backward()
    else:
        ref = m32(img, iters=iters)
        out = mbf(img.to(torch.bfloat16), iters=iters)
        ref.float().pow(2).mean().backward()
        out.float().pow(2).mean().backward()
    g32 = m32.image_to_tokens[1].weight.grad
    gbf = mbf.image_to_tokens[1].weight.grad
    print("g32 norm", g32.float().norm().item(), "gbf norm", gbf.float().norm().item())
    c = torch.nn.functional.cosine_similarity(g32.flatten().float(), gbf.flatten().float(), dim=0)
    print("cos:", c.item())
