"""Directed regression tests for the numerics quirks.

Each of these is an easy silent-divergence bug when re-implementing GLOM:
the -5e-4 self-mask (not -inf), top level /3, top-down zero-pad at the TOP
slot, k-only L2 normalization, pos-emb only into the top-down input, exact
(erf) GELU, trajectory includes initial state.
"""

import math

import torch
import torch.nn.functional as F

from glom_pytorch_amd import Glom, ConsensusAttention, GroupedFeedForward
from glom_pytorch_amd.models.glom import TOKEN_ATTEND_SELF_VALUE
from conftest import SMALL


def test_self_mask_is_mild_not_inf():
    """Diagonal gets -5e-4 — self still receives ~uniform attention weight."""
    att = ConsensusAttention(4, attend_self=False)
    lv = torch.randn(1, 16, 1, 8)
    q = lv
    k = F.normalize(lv, dim=-1)
    sim = torch.einsum("b i l d, b j l d -> b l i j", q, k) / math.sqrt(8)
    sim[0, 0].fill_diagonal_(TOKEN_ATTEND_SELF_VALUE)
    expect = sim.softmax(dim=-1)
    # self weight must be non-negligible (an -inf mask would zero it)
    out_attn_diag = expect[0, 0].diagonal()
    assert (out_attn_diag > 1e-3).all()
    out = att(lv)
    ref = torch.einsum("b l i j, b j l d -> b i l d", expect, lv)
    assert torch.allclose(out, ref, atol=1e-6)


def test_attend_self_true_no_mask():
    att = ConsensusAttention(4, attend_self=True)
    lv = torch.randn(1, 16, 1, 8)
    sim = torch.einsum("b i l d, b j l d -> b l i j",
                       lv, F.normalize(lv, dim=-1)) / math.sqrt(8)
    ref = torch.einsum("b l i j, b j l d -> b i l d", sim.softmax(-1), lv)
    assert torch.allclose(att(lv), ref, atol=1e-6)


def test_k_normalized_q_raw():
    """Only k is L2-normalized; q stays raw; scale is still d^-0.5."""
    att = ConsensusAttention(2, attend_self=True)
    lv = torch.randn(1, 4, 1, 8) * 7.0   # large magnitudes expose q-norm bugs
    out = att(lv)
    q = lv
    k = F.normalize(lv, dim=-1)
    sim = torch.einsum("b i l d, b j l d -> b l i j", q, k) / math.sqrt(8)
    ref = torch.einsum("b l i j, b j l d -> b i l d", sim.softmax(-1), lv)
    assert torch.allclose(out, ref, atol=1e-6)
    # sanity: normalizing q as well would give a DIFFERENT answer
    simqq = torch.einsum("b i l d, b j l d -> b l i j",
                         F.normalize(lv, dim=-1), k) / math.sqrt(8)
    refqq = torch.einsum("b l i j, b j l d -> b i l d", simqq.softmax(-1), lv)
    assert not torch.allclose(out, refqq, atol=1e-3)


def test_v_is_raw_levels():
    """Values are the raw levels, not the normalized ones."""
    att = ConsensusAttention(2, attend_self=True)
    lv = torch.randn(1, 4, 1, 8) * 5.0
    out = att(lv)
    # output rows are convex combinations of raw levels rows => can exceed
    # unit norm, which normalized-v output cannot
    assert out.norm(dim=-1).max() > 1.5


def test_top_level_divides_by_3():
    m = Glom(**SMALL)
    img = torch.randn(1, 3, 32, 32)
    tokens = m.image_to_tokens(img)
    b, n = 1, tokens.shape[1]
    lv = m.init_levels.view(1, 1, SMALL["levels"], SMALL["dim"]).expand(
        b, n, SMALL["levels"], SMALL["dim"])
    pos = m.pos_emb.weight[:n].view(1, n, 1, SMALL["dim"])
    bottom = tokens.view(b, n, 1, SMALL["dim"])
    out = m._eager_step(bottom, lv, pos)
    # recompute top level by hand with /3 (td contribution is exactly zero)
    bu_in = torch.cat((bottom, lv[..., :-1, :]), dim=-2)
    bu = m.bottom_up(bu_in)
    cons = m.attention(lv)
    top_ref = (lv[..., -1, :] + bu[..., -1, :] + cons[..., -1, :]) / 3.0
    assert torch.allclose(out[..., -1, :], top_ref, atol=1e-6)
    # and a /4 top level would be wrong
    assert not torch.allclose(out[..., -1, :],
                              (lv[..., -1, :] + bu[..., -1, :]
                               + cons[..., -1, :]) / 4.0, atol=1e-4)


def test_topdown_pad_is_at_top_slot():
    """top_down predicts levels 0..L-2; the TOP slot gets the zero pad."""
    m = Glom(**SMALL)
    torch.nn.init.zeros_(m.bottom_up.net[1].weight)
    torch.nn.init.zeros_(m.bottom_up.net[1].bias)
    torch.nn.init.zeros_(m.bottom_up.net[3].weight)
    torch.nn.init.constant_(m.bottom_up.net[3].bias, 0.0)
    # make top_down output a huge constant so its placement is visible
    torch.nn.init.zeros_(m.top_down.net[1].weight)
    torch.nn.init.zeros_(m.top_down.net[1].bias)
    torch.nn.init.zeros_(m.top_down.net[3].weight)
    torch.nn.init.constant_(m.top_down.net[3].bias, 100.0)
    img = torch.zeros(1, 3, 32, 32)
    lv0 = m(img, iters=0)
    lv1 = m(img, iters=1)
    L = SMALL["levels"]
    cons = m.attention(lv0)
    # lower levels see the +100 td; the top level must NOT
    for li in range(L - 1):
        ref = (lv0[..., li, :] + 0 + 100.0 + cons[..., li, :]) / 4.0
        assert torch.allclose(lv1[..., li, :], ref, atol=1e-4)
    ref_top = (lv0[..., L - 1, :] + 0 + cons[..., L - 1, :]) / 3.0
    assert torch.allclose(lv1[..., L - 1, :], ref_top, atol=1e-4)


def test_pos_emb_only_into_topdown():
    """Zeroing the top_down net must make pos_emb entirely irrelevant."""
    m = Glom(**SMALL)
    torch.nn.init.zeros_(m.top_down.net[1].weight)
    torch.nn.init.zeros_(m.top_down.net[1].bias)
    torch.nn.init.zeros_(m.top_down.net[3].weight)
    torch.nn.init.zeros_(m.top_down.net[3].bias)
    img = torch.randn(1, 3, 32, 32)
    out1 = m(img, iters=2)
    with torch.no_grad():
        m.pos_emb.weight.add_(torch.randn_like(m.pos_emb.weight))
    out2 = m(img, iters=2)
    assert torch.allclose(out1, out2, atol=1e-6)


def test_gelu_is_exact_erf():
    ff = GroupedFeedForward(dim=4, groups=1)
    x = torch.randn(1, 3, 1, 4)
    w1 = ff.net[1].weight[..., 0]
    b1 = ff.net[1].bias
    w2 = ff.net[3].weight[..., 0]
    b2 = ff.net[3].bias
    h = x[0, :, 0] @ w1.t() + b1
    exact = h * 0.5 * (1.0 + torch.erf(h / math.sqrt(2.0)))
    ref = exact @ w2.t() + b2
    assert torch.allclose(ff(x)[0, :, 0], ref, atol=1e-6)
    tanh_approx = F.gelu(h, approximate="tanh") @ w2.t() + b2
    assert not torch.equal(ff(x)[0, :, 0], tanh_approx)


def test_bottom_up_sees_tokens_as_level_minus_1():
    """Group 0 of bottom_up consumes the patch tokens, not level 0."""
    m = Glom(**SMALL)
    img = torch.randn(1, 3, 32, 32)
    tokens = m.image_to_tokens(img)
    n = tokens.shape[1]
    lv = m.init_levels.view(1, 1, SMALL["levels"], -1).expand(
        1, n, SMALL["levels"], SMALL["dim"])
    bu_in = torch.cat((tokens.unsqueeze(-2), lv[..., :-1, :]), dim=-2)
    bu = m.bottom_up(bu_in)
    # group 0 output computed directly from tokens
    d, mult = SMALL["dim"], 4
    w1 = m.bottom_up.net[1].weight[:d * mult, :, 0]
    b1 = m.bottom_up.net[1].bias[:d * mult]
    w2 = m.bottom_up.net[3].weight[:d, :, 0]
    b2 = m.bottom_up.net[3].bias[:d]
    ref0 = F.gelu(tokens @ w1.t() + b1) @ w2.t() + b2
    assert torch.allclose(bu[..., 0, :], ref0, atol=1e-5)


def test_token_grad_depth():
    """A top-level-only loss at trajectory index t reaches the patch tokens
    only when t >= levels: level grads shift down exactly one level per
    backward iteration through the bottom-up chain. (Found while validating
    the HIP engine: cos(0,0)=0 looked like a kernel bug but is semantics.)"""
    m = Glom(**SMALL)  # levels = 3
    img = torch.randn(1, 3, 32, 32)
    for t, expect_nonzero in ((2, False), (3, True)):
        m.zero_grad()
        out = m(img, iters=4, return_all=True)
        out[t, :, :, -1].pow(2).mean().backward()
        g = m.image_to_tokens[1].weight.grad
        assert (g.norm().item() > 0) == expect_nonzero, (t, g.norm().item())
