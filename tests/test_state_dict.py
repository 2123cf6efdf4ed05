"""state_dict layout contract + round-trip tests."""

import torch

from glom_pytorch_amd import Glom


def test_default_config_layout():
    m = Glom(dim=512, levels=6, image_size=224, patch_size=14)
    sd = m.state_dict()
    expect = {
        "init_levels": (6, 512),
        "image_to_tokens.1.weight": (512, 588),
        "image_to_tokens.1.bias": (512,),
        "pos_emb.weight": (256, 512),
        "bottom_up.net.1.weight": (12288, 512, 1),
        "bottom_up.net.1.bias": (12288,),
        "bottom_up.net.3.weight": (3072, 2048, 1),
        "bottom_up.net.3.bias": (3072,),
        "top_down.net.1.weight": (10240, 512, 1),
        "top_down.net.1.bias": (10240,),
        "top_down.net.3.weight": (2560, 2048, 1),
        "top_down.net.3.bias": (2560,),
    }
    assert set(sd.keys()) == set(expect.keys())
    for k, shape in expect.items():
        assert tuple(sd[k].shape) == shape, (k, sd[k].shape, shape)
    assert sum(p.numel() for p in m.parameters()) == 23_532_544


def test_radius_buffer_in_state_dict():
    m = Glom(dim=64, levels=3, image_size=32, patch_size=8,
             local_consensus_radius=2)
    sd = m.state_dict()
    assert "attention.non_local_mask" in sd
    assert tuple(sd["attention.non_local_mask"].shape) == (1, 16, 16)
    assert sd["attention.non_local_mask"].dtype == torch.bool


def test_roundtrip_strict():
    a = Glom(dim=64, levels=3, image_size=32, patch_size=8)
    b = Glom(dim=64, levels=3, image_size=32, patch_size=8)
    b.load_state_dict(a.state_dict(), strict=True)
    img = torch.randn(2, 3, 32, 32)
    assert torch.equal(a(img, iters=2), b(img, iters=2))
