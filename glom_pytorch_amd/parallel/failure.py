"""Minimal failure detection for fixed single-node DP over RCCL
(rank heartbeat via communicator health, abort on collective
timeout; no elastic membership needed for a fixed 8-GPU node)."""

from __future__ import annotations

import datetime
import os
import sys

import torch
import torch.distributed as dist


def init_distributed(timeout_s: int = 120) -> tuple[int, int]:
    """init_process_group with a hard collective timeout: a hung peer turns
    into a clean abort-and-relaunch instead of a wedged job. Returns
    (rank, world_size)."""
    if not dist.is_initialized():
        from glom_pytorch_amd.parallel.rccl_env import (
            apply_rccl_env_defaults)
        apply_rccl_env_defaults()
        dist.init_process_group(
            "nccl", timeout=datetime.timedelta(seconds=timeout_s))
    return dist.get_rank(), dist.get_world_size()


class Heartbeat:
    """Tiny periodic all-reduce proving every rank is alive and RCCL is
    healthy; raises (-> process exit -> launcher relaunch) on timeout."""

    def __init__(self, every_steps: int = 50, device=None):
        self.every = every_steps
        if device is None:
            device = (torch.device("cuda", torch.cuda.current_device())
                      if torch.cuda.is_available() else torch.device("cpu"))
        self.device = device
        self._buf = torch.ones(1, device=self.device)
        self._step = 0

    def tick(self):
        self._step += 1
        if self._step % self.every:
            return
        try:
            dist.all_reduce(self._buf)
            expect = float(dist.get_world_size())
            got = self._buf.item()
            self._buf.fill_(1.0)
            if got != expect:
                raise RuntimeError(
                    f"heartbeat mismatch: {got} != {expect} "
                    "(a rank died mid-collective)")
        except Exception as e:
            print(f"[rank {dist.get_rank()}] heartbeat failed: {e}",
                  file=sys.stderr, flush=True)
            self.abort()
            raise

    @staticmethod
    def abort():
        """Tear down the process group so the launcher can relaunch."""
        if dist.is_initialized():
            try:
                dist.destroy_process_group()
            except Exception:
                pass
        # non-zero exit signals torchrun to restart the job
        os.environ.setdefault("GLOM_ABORTED", "1")
