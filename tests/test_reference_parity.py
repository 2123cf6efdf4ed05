"""Cross-check against the upstream package when it is present on disk.

These tests run only on the dev container (where /root/reference is
mounted); they load the upstream module, copy state_dicts in both
directions and compare outputs. They are skipped anywhere the upstream
repo is absent (e.g. GPU boxes).
"""

import os
import sys

import pytest
import torch

from glom_pytorch_amd import Glom

REF = "/root/reference"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF, "glom_pytorch")),
    reason="upstream reference not mounted")


def _ref_glom(**kw):
    if REF not in sys.path:
        sys.path.insert(0, REF)
    from glom_pytorch import Glom as RefGlom
    return RefGlom(**kw)


@pytest.mark.parametrize("kw", [
    dict(dim=64, levels=3, image_size=32, patch_size=8),
    dict(dim=64, levels=3, image_size=32, patch_size=8, consensus_self=True),
    dict(dim=64, levels=3, image_size=32, patch_size=8,
         local_consensus_radius=2),
])
def test_forward_matches_upstream(kw):
    ours = Glom(**kw)
    ref = _ref_glom(**kw)
    ref.load_state_dict(ours.state_dict(), strict=True)
    img = torch.randn(2, 3, 32, 32)
    for it in (1, 3):
        a = ours(img, iters=it)
        b = ref(img, iters=it)
        assert torch.allclose(a, b, rtol=1e-5, atol=1e-6), \
            (a - b).abs().max().item()
    a = ours(img, iters=3, return_all=True)
    b = ref(img, iters=3, return_all=True)
    assert torch.allclose(a, b, rtol=1e-5, atol=1e-6)


def test_stateful_matches_upstream():
    kw = dict(dim=64, levels=3, image_size=32, patch_size=8)
    ours = Glom(**kw)
    ref = _ref_glom(**kw)
    ours.load_state_dict(ref.state_dict(), strict=True)  # reverse direction
    img1, img2 = torch.randn(2, 3, 32, 32), torch.randn(2, 3, 32, 32)
    la = ours(img2, iters=2, levels=ours(img1, iters=3))
    lb = ref(img2, iters=2, levels=ref(img1, iters=3))
    assert torch.allclose(la, lb, rtol=1e-5, atol=1e-6)


def test_gradients_match_upstream():
    kw = dict(dim=64, levels=3, image_size=32, patch_size=8)
    torch.manual_seed(3)
    ours = Glom(**kw)
    ref = _ref_glom(**kw)
    ref.load_state_dict(ours.state_dict(), strict=True)
    img = torch.randn(2, 3, 32, 32)

    out_a = ours(img, iters=3, return_all=True)
    out_b = ref(img, iters=3, return_all=True)
    # loss over the full trajectory reaches every parameter
    loss_a = out_a.pow(2).mean()
    loss_b = out_b.pow(2).mean()
    loss_a.backward()
    loss_b.backward()
    ref_named = dict(ref.named_parameters())
    for n, p in ours.named_parameters():
        q = ref_named[n]
        assert p.grad is not None and q.grad is not None, n
        assert torch.allclose(p.grad, q.grad, rtol=1e-4, atol=1e-7), \
            (n, (p.grad - q.grad).abs().max().item())
