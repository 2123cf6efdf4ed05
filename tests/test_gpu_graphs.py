"""hipGraph capture tests: graph replay must equal the eager-launched
native engine, re-capture on config change, and serve the stateful video
path (BASELINE config 5)."""

import pytest
import torch

from glom_pytorch_amd import Glom

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU"),
]

DEV = "cuda:0"
CFG = dict(dim=64, levels=3, image_size=32, patch_size=8)


def _model():
    torch.manual_seed(0)
    return Glom(**CFG).to(DEV, torch.bfloat16)


def test_graph_replay_matches_eager_launch():
    m = _model()
    img1 = torch.randn(2, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    img2 = torch.randn(2, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    with torch.no_grad():
        ref1 = m(img1, iters=3)
        ref2 = m(img2, iters=3)
        m.enable_graphs()
        out1 = m(img1, iters=3)   # capture + replay
        out2 = m(img2, iters=3)   # replay with new input
    assert torch.equal(out1, ref1)
    assert torch.equal(out2, ref2)   # not stale from capture input


def test_graph_recapture_on_iters_and_return_all():
    m = _model().enable_graphs()
    img = torch.randn(1, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    with torch.no_grad():
        a = m(img, iters=2)
        b = m(img, iters=4, return_all=True)
    assert a.shape == (1, 16, 3, 64)
    assert b.shape == (5, 1, 16, 3, 64)
    assert len(m._graph_cache.entries) == 2
    m.disable_graphs()
    with torch.no_grad():
        a2 = m(img, iters=2)
    assert torch.equal(a, a2)


def test_graph_stateful_video_path():
    """3-frame video: levels carried across calls with varying iters, each
    call its own captured graph (BASELINE config 5)."""
    m = _model()
    frames = [torch.randn(1, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
              for _ in range(3)]
    with torch.no_grad():
        l1 = m(frames[0], iters=6)
        l2 = m(frames[1], iters=4, levels=l1)
        l3 = m(frames[2], iters=2, levels=l2)
        m.enable_graphs()
        g1 = m(frames[0], iters=6)
        g2 = m(frames[1], iters=4, levels=g1)
        g3 = m(frames[2], iters=2, levels=g2)
    assert torch.equal(l1, g1)
    assert torch.equal(l2, g2)
    assert torch.equal(l3, g3)


def test_graph_training_unaffected():
    m = _model().enable_graphs()
    img = torch.randn(1, 3, 32, 32, device=DEV, dtype=torch.bfloat16)
    out = m(img, iters=2, return_all=True)   # grad enabled -> no graph
    loss = out[-1].float().pow(2).mean()
    loss.backward()
    assert m.init_levels.grad is not None
