"""HIP/CDNA4 op layer: loader for the in-tree gfx950 extension + autograd ops.

The extension (`_glom_hip*.so`) is built in-tree by ``setup.py build_ext
--inplace`` (PYTORCH_ROCM_ARCH=gfx950) so the binary travels with the repo
snapshot to GPU machines. On a GPU box the native path is mandatory: if a
bf16 CUDA tensor reaches ``native_forward`` and the extension cannot be
loaded, we raise instead of silently falling back to eager PyTorch.
"""

from __future__ import annotations

import glob
import importlib
import os

import torch

_EXT = None
_EXT_ERR: Exception | None = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    if _EXT_ERR is not None:
        raise _EXT_ERR
    try:
        from glom_pytorch_amd.ops import _glom_hip  # built in-tree
        _EXT = _glom_hip
        return _EXT
    except ImportError as e:
        here = os.path.dirname(__file__)
        sos = glob.glob(os.path.join(here, "_glom_hip*.so"))
        _EXT_ERR = ImportError(
            f"glom_pytorch_amd HIP extension not importable ({e}). "
            f"Found candidate .so files: {sos or 'none'}. Build it in-tree with "
            f"`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"from the repo root. The GPU path never falls back to eager.")
        raise _EXT_ERR


def available() -> bool:
    try:
        _load_extension()
        return True
    except ImportError:
        return False


def native_forward(model, img, iters, levels=None, return_all=False,
                   grad_iters=None, overlap_tail=False):
    """Run Glom.forward on the CDNA4 HIP engine (bf16, gfx950)."""
    from glom_pytorch_amd.ops.functional import glom_forward
    return glom_forward(model, img, iters=iters, levels=levels,
                        return_all=return_all, grad_iters=grad_iters,
                        overlap_tail=overlap_tail)
