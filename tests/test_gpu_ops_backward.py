"""Per-op backward parity on GPU: each custom Function's hand-written HIP
backward vs torch autograd on an fp32 replica of the same bf16 inputs."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU"),
]

DEV = "cuda:0"


def _rel(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / b.norm().clamp_min(1e-8)).item()


def _ff_ref(xs, w1, b1, w2, b2, G, d, m4):
    outs = []
    for g in range(G):
        h = xs[g] @ w1[g * m4:(g + 1) * m4].t() + b1[g * m4:(g + 1) * m4]
        y = F.gelu(h) @ w2[g * d:(g + 1) * d].t() + b2[g * d:(g + 1) * d]
        outs.append(y)
    return torch.stack(outs, dim=-2)


@pytest.mark.parametrize("mode", [0, 1])
def test_grouped_ff_backward(mode):
    from glom_pytorch_amd.ops.functional import GroupedFFFn
    torch.manual_seed(7)
    B, N, L, d = 2, 16, 3, 64
    m4 = 4 * d
    G = L if mode == 0 else L - 1
    bf = torch.bfloat16
    tokens = torch.randn(B, N, d, device=DEV, dtype=bf, requires_grad=True)
    levels = torch.randn(B, N, L, d, device=DEV, dtype=bf, requires_grad=True)
    pos = torch.randn(N, d, device=DEV, dtype=bf, requires_grad=True)
    w1 = (torch.randn(G * m4, d, device=DEV, dtype=bf) * 0.05).requires_grad_()
    b1 = torch.randn(G * m4, device=DEV, dtype=bf, requires_grad=True)
    w2 = (torch.randn(G * d, m4, device=DEV, dtype=bf) * 0.05).requires_grad_()
    b2 = torch.randn(G * d, device=DEV, dtype=bf, requires_grad=True)

    if mode == 0:
        out = GroupedFFFn.apply(tokens, levels, None, w1, b1, w2, b2, 0)
        inputs = (tokens, levels, w1, b1, w2, b2)
    else:
        out = GroupedFFFn.apply(None, levels, pos, w1, b1, w2, b2, 1)
        inputs = (levels, pos, w1, b1, w2, b2)
    gout = torch.randn_like(out)
    grads = torch.autograd.grad(out, inputs, gout)

    # fp32 autograd reference
    t32 = tokens.detach().float().requires_grad_()
    l32 = levels.detach().float().requires_grad_()
    p32 = pos.detach().float().requires_grad_()
    w1f = w1.detach().float().requires_grad_()
    b1f = b1.detach().float().requires_grad_()
    w2f = w2.detach().float().requires_grad_()
    b2f = b2.detach().float().requires_grad_()
    if mode == 0:
        xs = [t32] + [l32[..., g, :] for g in range(G - 1)]
        rinputs = (t32, l32, w1f, b1f, w2f, b2f)
    else:
        xs = [l32[..., g + 1, :] + p32.view(1, N, d) for g in range(G)]
        rinputs = (l32, p32, w1f, b1f, w2f, b2f)
    ref = _ff_ref(xs, w1f, b1f, w2f, b2f, G, d, m4)
    rgrads = torch.autograd.grad(ref, rinputs, gout.float(),
                                 allow_unused=True)

    names = (["tokens", "levels", "w1", "b1", "w2", "b2"] if mode == 0
             else ["levels", "pos", "w1", "b1", "w2", "b2"])
    for name, g, rg in zip(names, grads, rgrads):
        assert g is not None, name
        assert rg is not None, name
        r = _rel(g, rg)
        assert r < 5e-2, (name, r, g.float().norm().item(),
                          rg.float().norm().item())


def test_level_mix_backward():
    from glom_pytorch_amd.ops.functional import LevelMixFn
    torch.manual_seed(8)
    B, N, L, d = 2, 16, 3, 64
    bf = torch.bfloat16
    lv = torch.randn(B, N, L, d, device=DEV, dtype=bf, requires_grad=True)
    bu = torch.randn(B, N, L, d, device=DEV, dtype=bf, requires_grad=True)
    td = torch.randn(B, N, L - 1, d, device=DEV, dtype=bf, requires_grad=True)
    cons = torch.randn(B, N, L, d, device=DEV, dtype=bf, requires_grad=True)
    out = LevelMixFn.apply(lv, bu, td, cons)
    gout = torch.randn_like(out)
    g = torch.autograd.grad(out, (lv, bu, td, cons), gout)

    lvf, buf, tdf, consf = [t.detach().float().requires_grad_()
                            for t in (lv, bu, td, cons)]
    tdp = F.pad(tdf, (0, 0, 0, 1), value=0.0)
    c = torch.full((L,), 4.0, device=DEV)
    c[-1] = 3.0
    ref = (lvf + buf + tdp + consf) / c.view(1, 1, L, 1)
    rg = torch.autograd.grad(ref, (lvf, buf, tdf, consf), gout.float())
    for name, a, b in zip(["lv", "bu", "td", "cons"], g, rg):
        r = _rel(a, b)
        assert r < 2e-2, (name, r)


def test_full_graph_token_grad_nonzero():
    """The patch-token gradient path through every iteration must be live."""
    from glom_pytorch_amd import Glom
    torch.manual_seed(0)
    m = Glom(dim=64, levels=3, image_size=32, patch_size=8)
    m = m.to(DEV, torch.bfloat16)
    img = torch.randn(2, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    out = m(img, iters=3, return_all=True)
    loss = out[-1].float().pow(2).mean()
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None, n
        assert p.grad.float().norm().item() > 0, n
