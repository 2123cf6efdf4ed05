"""Tracing / profiling / metrics utilities.

- roctx ranges around the iteration loop and each engine op (rocprofv3
  picks them up with --marker-trace; torch.cuda.nvtx maps to roctx on ROCm)
- step timing via hip events + achieved-TFLOPs accounting from the GLOM
  FLOP model
- JSONL metrics sink
"""

from __future__ import annotations

import contextlib
import json
import time

import torch


@contextlib.contextmanager
def trace_range(name: str):
    """roctx range (no-op overhead when no profiler is attached)."""
    if torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def glom_flops(batch: int, *, dim: int, levels: int, image_size: int,
               patch_size: int, iters: int, backward: bool = True) -> float:
    """FLOPs of one GLOM forward (x3 with backward), from SURVEY.md §2.3."""
    side = image_size // patch_size
    n = side * side
    m4 = 4 * dim
    up = 2 * n * dim * m4
    ff = (levels + (levels - 1)) * 2 * up           # bottom-up + top-down
    attn = 2 * levels * (2 * n * n * dim)           # scores + AV
    embed = 2 * n * (patch_size ** 2 * 3) * dim
    total = batch * (iters * (ff + attn) + embed)
    return total * (3.0 if backward else 1.0)


class StepTimer:
    """CUDA-event step timer with rolling stats."""

    def __init__(self):
        self._start = None
        self.times_ms: list[float] = []

    def start(self):
        self._start = torch.cuda.Event(enable_timing=True)
        self._end = torch.cuda.Event(enable_timing=True)
        self._start.record()

    def stop(self) -> float:
        self._end.record()
        self._end.synchronize()
        ms = self._start.elapsed_time(self._end)
        self.times_ms.append(ms)
        return ms

    @property
    def mean_ms(self):
        return sum(self.times_ms) / max(1, len(self.times_ms))


class MetricsLogger:
    """stdout + JSONL metrics (images/sec, step time, loss, TFLOPs)."""

    def __init__(self, path: str | None = None, stdout: bool = True):
        self.path = path
        self.stdout = stdout

    def log(self, step: int, **metrics):
        rec = {"step": step, "time": time.time(), **metrics}
        if self.path:
            with open(self.path, "a") as f:
                f.write(json.dumps(rec) + "\n")
        if self.stdout:
            parts = " ".join(f"{k}={v:.4g}" if isinstance(v, float)
                             else f"{k}={v}" for k, v in metrics.items())
            print(f"[step {step}] {parts}", flush=True)
