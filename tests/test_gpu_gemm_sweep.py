"""Dispatch-coverage sweep: drive every GEMM kernel variant (v1 predicated,
nt_fast 128^2, nt_fast3/4 128x256, tn_fast(+split-K), nn_fast) through the
grouped-FF and consensus ops across shapes, checking against fp32 torch on
the same bf16 inputs."""

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


# (B, N, L, d): chosen to hit v1 tails, nt_fast, nt3/nt4, split-K branches
SHAPES = [
    (2, 16, 3, 64),      # tiny: v1 predicated everywhere
    (8, 16, 3, 128),     # M=128: nt_fast (128^2), m4=512
    (2, 64, 2, 256),     # M=128, m4=1024: nt4 up, K=256
    (4, 256, 3, 64),     # M=1024, m4=256: nt_fast + tn split-K (K=1024)
    (1, 1024, 2, 128),   # M=1024, N(attn)=1024: big-N consensus fallback
    (16, 256, 2, 512),   # M=4096: headline-like, nt4 + tn_sk + fused attn
]


@pytest.mark.parametrize("B,N,L,d", SHAPES)
def test_grouped_ff_fwd_bwd_shapes(B, N, L, d):
    from glom_pytorch_amd.ops.functional import GroupedFFFn
    torch.manual_seed(B * 1000 + N)
    m4 = 4 * d
    bf = torch.bfloat16
    tokens = torch.randn(B, N, d, device=DEV, dtype=bf, requires_grad=True)
    levels = torch.randn(B, N, L, d, device=DEV, dtype=bf,
                         requires_grad=True)
    w1 = (torch.randn(L * m4, d, device=DEV, dtype=bf) * 0.04).requires_grad_()
    b1 = torch.randn(L * m4, device=DEV, dtype=bf, requires_grad=True)
    w2 = (torch.randn(L * d, m4, device=DEV, dtype=bf) * 0.04).requires_grad_()
    b2 = torch.randn(L * d, device=DEV, dtype=bf, requires_grad=True)

    out = GroupedFFFn.apply(tokens, levels, None, w1, b1, w2, b2, 0)
    gout = torch.randn_like(out)
    grads = torch.autograd.grad(out, (tokens, levels, w1, b1, w2, b2), gout)

    xs = [tokens.detach().float()] + [levels.detach().float()[..., g, :]
                                      for g in range(L - 1)]
    t32 = xs[0].requires_grad_()
    l32 = levels.detach().float().requires_grad_()
    xs = [t32] + [l32[..., g, :] for g in range(L - 1)]
    wf = [w.detach().float().requires_grad_() for w in (w1, b1, w2, b2)]
    outs = []
    for g in range(L):
        h = xs[g] @ wf[0][g * m4:(g + 1) * m4].t() + wf[1][g * m4:(g + 1) * m4]
        outs.append(F.gelu(h) @ wf[2][g * d:(g + 1) * d].t()
                    + wf[3][g * d:(g + 1) * d])
    ref = torch.stack(outs, dim=-2)
    rgrads = torch.autograd.grad(ref, (t32, l32, *wf), gout.float(),
                                 allow_unused=True)

    assert _rel(out, ref) < 1.5e-2
    names = ["tokens", "levels", "w1", "b1", "w2", "b2"]
    for n, g, rg in zip(names, grads, rgrads):
        assert _rel(g, rg) < 6e-2, (n, _rel(g, rg))


@pytest.mark.parametrize("B,N,L,d", [(2, 64, 2, 256), (1, 1024, 2, 128),
                                     (4, 256, 2, 512)])
def test_consensus_fwd_bwd_shapes(B, N, L, d):
    import math
    from glom_pytorch_amd.ops.functional import ConsensusFn
    torch.manual_seed(N + d)
    lv = torch.randn(B, N, L, d, device=DEV, dtype=torch.bfloat16,
                     requires_grad=True)
    out = ConsensusFn.apply(lv, False, None)
    gout = torch.randn_like(out)
    (dlev,) = torch.autograd.grad(out, lv, gout)

    x = lv.detach().float().requires_grad_()
    sim = torch.einsum("bild,bjld->blij", x,
                       F.normalize(x, dim=-1)) / math.sqrt(d)
    eye = torch.eye(N, device=DEV, dtype=torch.bool)
    sim = sim.masked_fill(eye.view(1, 1, N, N), -5e-4)
    ref = torch.einsum("blij,bjld->bild", sim.softmax(-1), x)
    (dref,) = torch.autograd.grad(ref, x, gout.float())
    assert _rel(out, ref) < 1.5e-2
    assert _rel(dlev, dref) < 5e-2


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_nt5p_dispatches_on_headline_shapes():
    """The persistent-ring kernel must actually own the up/dH shapes —
    guards against a dispatcher regression silently rerouting to nt4."""
    import os
    import subprocess
    import sys
    env = dict(os.environ, GLOM_DISPATCH_DEBUG="1")
    code = ("import torch; from glom_pytorch_amd.ops import _load_extension;"
            " ext=_load_extension(); ext.bench_gemm(16384,2048,512,0,6,0,1)")
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    assert "nt5p=1" in r.stderr, r.stderr[-800:]


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_nt8p_dispatches_on_down_projection():
    """The 8-phase 256^2 kernel must own the K>=2048 NT shapes (default
    on since the overlap schedule flipped its e2e sign) — and must NOT
    take the K=512 up-projection family."""
    import os
    import subprocess
    import sys
    env = dict(os.environ, GLOM_DISPATCH_DEBUG="1")
    code = ("import torch; from glom_pytorch_amd.ops import _load_extension;"
            " ext=_load_extension();"
            " ext.bench_gemm(16384,512,2048,0,6,0,1);"     # down-proj
            " ext.bench_gemm(16384,2048,512,0,6,0,1)")     # up-proj
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    lines = [l for l in r.stderr.splitlines() if "[dispatch]" in l]
    down = [l for l in lines if "N512 K2048" in l]
    up = [l for l in lines if "N2048 K512" in l]
    assert down and "nt8p=1" in down[0], down
    assert up and "nt8p=0" in up[0] and "nt5p=1" in up[0], up
