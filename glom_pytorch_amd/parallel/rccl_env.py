"""RCCL environment presets for the single-node 8x MI355X topology.

Each MI355X has 7 point-to-point xGMI links (~153 GB/s each) to the other
7 GPUs — a fully-connected graph, not a switch. A ring all-reduce is
per-link bound (one link in + one out per GPU per ring), so engaging all
7 links needs multiple concurrent rings/channels: RCCL builds one ring
per channel, rotating the link assignment, and ~4 channels per ring
direction saturate a link. 28 channels (7 links x 4) is the preset that
covers every link without burning excessive CUs on copy kernels.

All values are `setdefault`: anything the user or launcher exports wins.
Call before `init_process_group` (RCCL reads env at communicator init).
"""

from __future__ import annotations

import os

RCCL_DEFAULTS = {
    # one ring per channel; 28 = 7 xGMI links x 4 channels each, enough
    # concurrent rings to engage every point-to-point link
    "NCCL_MIN_NCHANNELS": "28",
    # single node: never try network transports
    "NCCL_IB_DISABLE": "1",
    # dmabuf IPC is the only mode the host driver supports (container
    # contract; exported by the image, restated here for safety)
    "HSA_ENABLE_IPC_MODE_LEGACY": "0",
}


def apply_rccl_env_defaults() -> dict:
    """Apply the presets (without clobbering user overrides) and return
    the effective values for logging."""
    eff = {}
    for k, v in RCCL_DEFAULTS.items():
        os.environ.setdefault(k, v)
        eff[k] = os.environ[k]
    return eff
