"""Round-2 GPU tests: native patch embedding (K1), fused AdamW, the
hipGraph-captured training step, and the headline-config direct parity
assert."""

import pytest
import torch

from glom_pytorch_amd import Glom

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU"),
]

DEV = "cuda:0"


def _rel_err(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / b.norm().clamp_min(1e-12)).item()


def _cos(a, b):
    a, b = a.float().flatten(), b.float().flatten()
    return torch.nn.functional.cosine_similarity(a, b, dim=0).item()


# ------------------------- patch embedding (K1) ------------------------ #

@pytest.mark.parametrize("dim,patch,size,batch", [
    (512, 14, 224, 2),     # headline shape (588 -> padded 640)
    (64, 8, 32, 3),        # plumbing shape (192, already /64)
])
def test_patch_embed_parity(dim, patch, size, batch):
    from glom_pytorch_amd.ops.functional import PatchEmbedFn
    torch.manual_seed(0)
    w = (torch.randn(dim, patch * patch * 3, device=DEV) * 0.05) \
        .to(torch.bfloat16)
    b = torch.randn(dim, device=DEV).to(torch.bfloat16) * 0.1
    img = torch.randn(batch, 3, size, size, device=DEV).to(torch.bfloat16)

    wg = w.clone().requires_grad_(True)
    bg = b.clone().requires_grad_(True)
    ig = img.clone().requires_grad_(True)
    out = PatchEmbedFn.apply(ig, wg, bg, patch)

    # fp32 reference on the same bf16 inputs
    from einops import rearrange
    x32 = rearrange(img.float(), "b c (h p1) (w p2) -> b (h w) (p1 p2 c)",
                    p1=patch, p2=patch)
    w32 = w.float().requires_grad_(True)
    b32 = b.float().requires_grad_(True)
    x32 = x32.requires_grad_(True)
    ref = x32 @ w32.t() + b32
    assert _rel_err(out, ref) < 2e-2

    dY = torch.randn_like(ref)
    ref.backward(dY)
    out.backward(dY.to(torch.bfloat16))
    assert _rel_err(wg.grad, w32.grad) < 2e-2
    assert _rel_err(bg.grad, b32.grad) < 2e-2
    # image grad goes through the unpatchify scatter
    dimg_ref = rearrange(x32.grad, "b (h w) (p1 p2 c) -> b c (h p1) (w p2)",
                         h=size // patch, w=size // patch, p1=patch,
                         p2=patch)
    assert _rel_err(ig.grad, dimg_ref) < 2e-2


def test_patch_embed_in_model_trace():
    """The native forward must route K1 through the HIP patch embed (no
    rocBLAS nn.Linear)."""
    torch.manual_seed(0)
    m = Glom(dim=64, levels=3, image_size=32, patch_size=8).to(
        DEV, torch.bfloat16)
    img = torch.randn(2, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    out = m(img, iters=2)
    assert torch.isfinite(out.float()).all()
    # grad_fn chain of tokens goes through PatchEmbedFn
    img.requires_grad_(True)
    out = m(img, iters=1)
    out.sum().backward()
    assert img.grad is not None
    assert torch.isfinite(img.grad.float()).all()


# ---------------------------- fused AdamW ------------------------------ #

def _torch_adamw_reference(masters, grads_seq, lr, clip):
    opt = torch.optim.AdamW(masters, lr=lr, foreach=True)
    for grads in grads_seq:
        for m, g in zip(masters, grads):
            m.grad = g.float()
        if clip:
            torch.nn.utils.clip_grad_norm_(masters, clip)
        opt.step()
        opt.zero_grad(set_to_none=True)
    return masters, opt


@pytest.mark.parametrize("clip", [1.0, 0.0])
def test_fused_adamw_matches_torch(clip):
    from glom_pytorch_amd.ops.optim import FusedAdamW
    torch.manual_seed(0)
    shapes = [(512, 17), (3,), (128,), (64, 9, 5)]
    base = [torch.randn(s, device=DEV) for s in shapes]
    params = [b.to(torch.bfloat16) for b in base]
    masters = [p.float() for p in params]
    ref_masters = [m.clone() for m in masters]

    steps = 5
    grads_seq = [[(torch.randn(s, device=DEV) * 3).to(torch.bfloat16)
                  for s in shapes] for _ in range(steps)]

    fused = FusedAdamW(params, masters, lr=1e-2,
                       max_grad_norm=clip)
    for grads in grads_seq:
        for p, g in zip(params, grads):
            p.grad = g
        fused.step()

    ref, _ = _torch_adamw_reference(ref_masters, grads_seq, 1e-2, clip)
    for m, r in zip(masters, ref):
        assert _rel_err(m, r) < 1e-4, (m.shape, _rel_err(m, r))
    for p, r in zip(params, ref):
        assert _rel_err(p, r.to(torch.bfloat16)) < 1e-2
    assert fused.steps_done() == steps


def test_fused_adamw_state_dict_interchange():
    from glom_pytorch_amd.ops.optim import FusedAdamW
    torch.manual_seed(1)
    params = [torch.randn(16, 8, device=DEV).to(torch.bfloat16)
              for _ in range(2)]
    masters = [p.float() for p in params]
    fused = FusedAdamW(params, masters, lr=3e-3, max_grad_norm=1.0)
    for _ in range(3):
        for p in params:
            p.grad = torch.randn_like(p)
        fused.step()
    sd = fused.state_dict()
    # load into a stock torch.optim.AdamW over clones of the masters
    clones = [m.clone().requires_grad_(True) for m in masters]
    topt = torch.optim.AdamW(clones, lr=3e-3)
    topt.load_state_dict(sd)
    # and back
    fused2 = FusedAdamW(params, [m.clone() for m in masters], lr=3e-3)
    fused2.load_state_dict(topt.state_dict())
    for a, b in zip(fused.exp_avg, fused2.exp_avg):
        assert torch.allclose(a, b)
    assert fused2.steps_done() == 3


# ----------------------- graphed training step ------------------------- #

def _trainer(graph, seed=0, **kw):
    from glom_pytorch_amd.parallel.trainer import DenoisingTrainer
    torch.manual_seed(seed)
    m = Glom(dim=64, levels=3, image_size=32, patch_size=8).to(
        DEV, torch.bfloat16)
    return DenoisingTrainer(m, lr=1e-3, noise_std=0.0, graph_step=graph,
                            **kw)


def test_graph_step_matches_eager():
    """The captured step (noise + fwd + bwd + fused AdamW in one hipGraph)
    must track the eager-launched step. noise_std=0 removes RNG."""
    img = torch.randn(4, 3, 32, 32, device=DEV).to(torch.bfloat16)

    te = _trainer(graph=False)
    tg = _trainer(graph=True)
    # identical init
    tg.model.load_state_dict(te.model.state_dict())
    tg.decoder.load_state_dict(te.decoder.state_dict())
    with torch.no_grad():
        for mw, q in zip(tg.master, tg._params):
            mw.copy_(q.float())

    losses_e = [te.step(img, iters=3) for _ in range(4)]
    losses_g = [tg.step(img, iters=3) for _ in range(4)]
    assert len(tg._graphs) == 1 and tg._graphs != {}
    entry = next(iter(tg._graphs.values()))
    assert entry is not False, "capture fell back to eager"
    for le, lg in zip(losses_e, losses_g):
        assert abs(le - lg) < 5e-2 * max(1.0, abs(le)), (le, lg)
    for (ne, pe), (ng, pg) in zip(te.model.named_parameters(),
                                  tg.model.named_parameters()):
        assert _cos(pe, pg) > 0.999, (ne, _cos(pe, pg))


def test_graph_step_soak_and_recapture():
    img = torch.randn(4, 3, 32, 32, device=DEV).to(torch.bfloat16)
    tg = _trainer(graph=True, seed=2)
    for _ in range(10):
        loss = tg.step(img, iters=3, sync_loss=False)
    assert torch.isfinite(loss).all()
    # new iters -> second capture
    tg.step(img, iters=2)
    assert len(tg._graphs) == 2


# ------------------- headline-config direct parity --------------------- #

def test_headline_config_direct_parity():
    """dim=512 L=6 224/14 — the exact shipping shape, asserted end-to-end:
    bf16 native fwd+bwd vs fp32 eager on identical weights/inputs
    (VERDICT.md round-1 weak item 5)."""
    torch.manual_seed(0)
    cfg = dict(dim=512, levels=6, image_size=224, patch_size=14)
    m32 = Glom(**cfg).to(DEV)
    m32.force_eager = True
    mbf = Glom(**cfg).to(DEV)
    mbf.load_state_dict(m32.state_dict())
    mbf = mbf.to(torch.bfloat16)

    img = torch.randn(2, 3, 224, 224, device=DEV)
    ref = m32(img, iters=3, return_all=True)
    out = mbf(img.to(torch.bfloat16), iters=3, return_all=True)
    assert out.shape == (4, 2, 256, 6, 512)
    assert _rel_err(out, ref) < 2e-2, _rel_err(out, ref)

    # loss over the whole trajectory (skipping the initial state): touches
    # every (time, level) so every parameter gets a structurally nonzero
    # gradient at iters=3. (A top-level-only loss at t<levels would give
    # the patch embed an EXACTLY zero grad: bottom-up information climbs
    # one level per iteration — that is faithful reference semantics.)
    ref[1:].float().pow(2).mean().backward()
    out[1:].float().pow(2).mean().backward()
    for (n32, p32), (nbf, pbf) in zip(m32.named_parameters(),
                                      mbf.named_parameters()):
        assert p32.grad is not None and pbf.grad is not None, n32
        c = _cos(p32.grad, pbf.grad)
        assert c > 0.99, (n32, c)


# ------------- consensus attention beyond the fused N=256 -------------- #

@pytest.mark.parametrize("side,dim", [
    (24, 64),     # N=576: not a multiple of 256 -> generic softmax path
    (32, 64),     # N=1024: stretch-config grid
])
def test_consensus_parity_large_n(side, dim):
    """The fused softmax/AV kernel serves N==256; every other patch-grid
    size takes the scores->masked-softmax->AV path. Pin fwd+bwd parity on
    those shapes directly (VERDICT round-1 weak item 4: the N==256
    special case must not be a correctness cliff)."""
    from glom_pytorch_amd.ops.functional import ConsensusFn
    torch.manual_seed(0)
    N, L, B = side * side, 2, 1
    lv = (torch.randn(B, N, L, dim, device=DEV) * 0.5).to(torch.bfloat16)

    for attend_self in (False, True):
        lg = lv.clone().requires_grad_(True)
        out = ConsensusFn.apply(lg, attend_self, None)

        l32 = lv.float().requires_grad_(True)
        q = l32
        k = torch.nn.functional.normalize(l32, dim=-1)
        sim = torch.einsum("bild,bjld->blij", q, k) * dim ** -0.5
        if not attend_self:
            eye = torch.eye(N, device=DEV, dtype=torch.bool)
            sim = sim.masked_fill(eye.view(1, 1, N, N), -5e-4)
        attn = sim.softmax(dim=-1)
        ref = torch.einsum("blij,bjld->bild", attn, l32)
        assert _rel_err(out, ref) < 2e-2, (side, attend_self)

        dO = torch.randn_like(ref)
        ref.backward(dO)
        out.backward(dO.to(torch.bfloat16))
        assert _rel_err(lg.grad, l32.grad) < 3e-2, (side, attend_self)


def test_overlap_tail_matches_sequential():
    """Running the forward-only tail concurrently with the backward must
    not change training: compare two trainers (overlap on/off), same
    init, noise_std=0, several steps."""
    img = torch.randn(4, 3, 32, 32, device=DEV).to(torch.bfloat16)
    ta = _trainer(graph=False, overlap_tail=False)
    tb = _trainer(graph=False, overlap_tail=True)
    tb.model.load_state_dict(ta.model.state_dict())
    tb.decoder.load_state_dict(ta.decoder.state_dict())
    with torch.no_grad():
        for mw, q in zip(tb.master, tb._params):
            mw.copy_(q.float())
    la = [ta.step(img, iters=4) for _ in range(3)]
    lb = [tb.step(img, iters=4) for _ in range(3)]
    for a, b in zip(la, lb):
        assert abs(a - b) < 1e-3 * max(1.0, abs(a)), (a, b)
    for (na, pa), (nb, pb) in zip(ta.model.named_parameters(),
                                  tb.model.named_parameters()):
        assert torch.equal(pa, pb), na


def test_micro_pipeline_matches_full_batch():
    """Pipelined microbatching (chunk i+1 forward overlapping chunk i
    backward, losses scaled 1/m) must produce the full-batch gradients:
    mean-MSE over the batch equals the mean of the chunk means."""
    img = torch.randn(8, 3, 32, 32, device=DEV).to(torch.bfloat16)
    tf = _trainer(graph=False)
    tm = _trainer(graph=False, micro_batches=2)
    tm.model.load_state_dict(tf.model.state_dict())
    tm.decoder.load_state_dict(tf.decoder.state_dict())
    with torch.no_grad():
        for mw, q in zip(tm.master, tm._params):
            mw.copy_(q.float())
    lf = [tf.step(img, iters=4) for _ in range(3)]
    lm = [tm.step(img, iters=4) for _ in range(3)]
    for a, b in zip(lf, lm):
        assert abs(a - b) < 2e-2 * max(1.0, abs(a)), (a, b)
    for (nf, pf), (nm, pm) in zip(tf.model.named_parameters(),
                                  tm.model.named_parameters()):
        assert _cos(pf, pm) > 0.999, (nf, _cos(pf, pm))
