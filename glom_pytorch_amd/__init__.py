"""glom_pytorch_amd — an MI355X-native GLOM engine.

A from-scratch reimplementation of the capabilities of lucidrains/glom-pytorch
(reference: /root/reference/glom_pytorch/glom_pytorch.py) built MI355X-first:

- ``Glom`` keeps the reference's public API and state_dict layout byte-for-byte
  (reference glom_pytorch.py:78-150), so checkpoints interchange freely.
- On a gfx950 GPU with bf16 tensors, the per-iteration hot path (grouped
  bottom-up/top-down MLPs, consensus attention, level mixing) runs on
  hand-written CDNA4 HIP kernels (MFMA dense tiles + LDS staging) exposed via
  ``glom_pytorch_amd.ops``.
- The T-step iteration loop can be hipGraph-captured (``glom_pytorch_amd.graphs``).
- Data-parallel denoising training over RCCL/xGMI lives in
  ``glom_pytorch_amd.parallel``.
"""

from glom_pytorch_amd.models.glom import Glom, GroupedFeedForward, ConsensusAttention

__version__ = "0.2.0"

__all__ = ["Glom", "GroupedFeedForward", "ConsensusAttention"]
