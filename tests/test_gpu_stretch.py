"""Stretch configuration (BASELINE config 4): dim=1024, levels=12,
image 512/16 -> N=1024 patch columns. Exercises the L=12 group tables,
N=1024 attention rows, and the big-shape GEMM dispatch."""

import pytest
import torch

from glom_pytorch_amd import Glom

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU"),
]

DEV = "cuda:0"


def test_stretch_fwd_bwd():
    torch.manual_seed(0)
    m = Glom(dim=1024, levels=12, image_size=512, patch_size=16)
    m = m.to(DEV, torch.bfloat16)
    img = torch.randn(2, 3, 512, 512, device=DEV, dtype=torch.bfloat16)
    out = m(img, iters=4, return_all=True)
    assert out.shape == (5, 2, 1024, 12, 1024)
    loss = out[-1].float().pow(2).mean()
    loss.backward()
    for n, p in m.named_parameters():
        assert torch.isfinite(p.grad.float()).all(), n
    torch.cuda.synchronize()


def test_stretch_parity_small_iters():
    """Numerical parity vs eager fp32 at the stretch dims (scaled-down
    batch/iters to keep runtime sane)."""
    torch.manual_seed(0)
    m32 = Glom(dim=1024, levels=12, image_size=512, patch_size=16).to(DEV)
    m32.force_eager = True
    mbf = Glom(dim=1024, levels=12, image_size=512, patch_size=16).to(DEV)
    mbf.load_state_dict(m32.state_dict())
    mbf = mbf.to(torch.bfloat16)
    img = torch.randn(1, 3, 512, 512, device=DEV)
    with torch.no_grad():
        ref = m32(img, iters=2)
        out = mbf(img.to(torch.bfloat16), iters=2)
    rel = ((out.float() - ref).norm() / ref.norm()).item()
    assert rel < 2e-2, rel
