import torch, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd import Glom
from glom_pytorch_amd.ops import functional as Fn

DEV = "cuda:0"
torch.manual_seed(0)

def run(alias):
    torch.manual_seed(0)
    m = Glom(dim=64, levels=3, image_size=32, patch_size=8).to(DEV, torch.bfloat16)
    if not alias:
        orig = Fn.LevelMixFn.backward
        class NoAlias(torch.autograd.Function):
            @staticmethod
            def forward(ctx, levels, bu, td, cons):
                from glom_pytorch_amd.ops import _load_extension
                return _load_extension().level_mix_fwd(levels, bu, td, cons)
            @staticmethod
            def backward(ctx, dout):
                from glom_pytorch_amd.ops import _load_extension
                dmix, dtd = _load_extension().level_mix_bwd(dout.contiguous())
                return dmix, dmix.clone(), dtd, dmix.clone()
        saved = Fn.LevelMixFn
        Fn.LevelMixFn = NoAlias
    img = torch.randn(2, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    out = m(img, iters=3, return_all=True)
    loss = out[2, :, :, -1].float().pow(2).mean()
    loss.backward()
    print(f"alias={alias}")
    for n, p in m.named_parameters():
        print(f"  {n:35s} grad_norm={p.grad.float().norm().item():.6e}")
    if not alias:
        Fn.LevelMixFn = saved

run(alias=True)
run(alias=False)
