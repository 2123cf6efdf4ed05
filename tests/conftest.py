import os
import sys

import pytest
import torch

# Make the repo root importable regardless of where pytest is invoked from.
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD MI355X (gfx950) GPU")


@pytest.fixture(autouse=True)
def _deterministic_seed():
    torch.manual_seed(0)


# Small plumbing config from BASELINE.json: dim=64 L=3 32/8 iters=3 batch=2.
SMALL = dict(dim=64, levels=3, image_size=32, patch_size=8)
SMALL_ITERS = 3
SMALL_BATCH = 2
