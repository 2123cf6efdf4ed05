"""Utilities: FLOP model, metrics sink, trace ranges (CPU-safe no-ops)."""

import json

import torch

from glom_pytorch_amd.utils.profiling import (MetricsLogger, glom_flops,
                                              trace_range)


def test_glom_flops_model():
    # headline config: ~151.4 GFLOP forward at B=1 iters=12
    f = glom_flops(1, dim=512, levels=6, image_size=224, patch_size=14,
                   iters=12, backward=False)
    assert 140e9 < f < 165e9, f
    # backward multiplies by 3, batch is linear
    fb = glom_flops(4, dim=512, levels=6, image_size=224, patch_size=14,
                    iters=12, backward=True)
    assert abs(fb - 12 * f) / (12 * f) < 1e-9


def test_metrics_logger_jsonl(tmp_path):
    p = str(tmp_path / "m.jsonl")
    ml = MetricsLogger(p, stdout=False)
    ml.log(1, loss=0.5, images_sec=100.0)
    ml.log(2, loss=0.25)
    recs = [json.loads(l) for l in open(p)]
    assert [r["step"] for r in recs] == [1, 2]
    assert recs[0]["images_sec"] == 100.0


def test_trace_range_noop_on_cpu():
    with trace_range("x"):
        y = torch.ones(2) + 1
    assert y.sum() == 4
