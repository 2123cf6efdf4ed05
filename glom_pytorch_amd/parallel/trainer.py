"""Denoising self-supervised trainer for GLOM (the reference's training
recipe, README.md:56-90): noise the image, run the model with return_all,
decode the top level at a mid-iteration timestep back to pixels, MSE against
the clean image. Adds what the reference leaves to the user: optimizer,
data-parallel gradient sync, checkpointing, metrics."""

from __future__ import annotations

import json
import os
import time

import torch
import torch.distributed as dist
import torch.nn.functional as F
from einops.layers.torch import Rearrange
from torch import nn

from glom_pytorch_amd.parallel.ddp import BucketedDDP
from glom_pytorch_amd.parallel.failure import Heartbeat


class DenoisingDecoder(nn.Sequential):
    """Top-level embedding -> image patches (README.md:78-81)."""

    def __init__(self, dim: int, image_size: int, patch_size: int):
        side = image_size // patch_size
        super().__init__(
            nn.Linear(dim, patch_size ** 2 * 3),
            Rearrange("b (h w) (p1 p2 c) -> b c (h p1) (w p2)",
                      h=side, w=side, p1=patch_size, p2=patch_size),
        )


class DenoisingTrainer:
    def __init__(self, model, *, lr=3e-4, noise_std=1.0, decode_step=7,
                 grad_clip=1.0, distributed=False, bucket_bytes=16 << 20,
                 process_group=None, log_path=None):
        """``process_group``: the group whose ranks are data-parallel
        replicas (default: the global group). For 2-D DP x SP meshes pass
        the DP subgroup here — BucketedDDP AVERAGES over its group, which
        is correct for DP replicas; SP shards need a separate SUM
        all-reduce over the SP subgroup (see parallel/sequence.py)."""
        self.model = model
        self.dim = model.dim
        self.decode_step = decode_step
        self.noise_std = noise_std
        p = next(model.parameters())
        self.decoder = DenoisingDecoder(
            model.dim, model.image_size, model.patch_size
        ).to(p.device, p.dtype)
        self.distributed = distributed and dist.is_initialized()
        self.ddp_model = self.ddp_dec = self.heartbeat = None
        if self.distributed:
            self.ddp_model = BucketedDDP(model, bucket_bytes,
                                         process_group=process_group)
            self.ddp_dec = BucketedDDP(self.decoder, bucket_bytes,
                                       process_group=process_group)
            # rank-liveness probe: a hung peer becomes a clean abort
            self.heartbeat = Heartbeat(every_steps=50)
        params = list(model.parameters()) + list(self.decoder.parameters())
        # bf16 training keeps fp32 master weights: updates of size lr*grad
        # would otherwise partially round away in bf16 parameter storage
        self.master = None
        if p.dtype == torch.bfloat16:
            self.master = [q.detach().float().requires_grad_(False)
                           for q in params]
            self._params = params
            self.opt = torch.optim.AdamW(self.master, lr=lr, foreach=True)
        else:
            self.opt = torch.optim.AdamW(params, lr=lr, foreach=True)
        self.grad_clip = grad_clip
        self.step_idx = 0
        self.log_path = log_path

    def step(self, img: torch.Tensor, iters: int | None = None,
             sync_loss: bool = True):
        """One denoising training step. With sync_loss=False the loss is
        returned as a device tensor (no host sync), letting the host run
        ahead and queue the next step's launches."""
        iters = iters if iters is not None else 2 * self.model.levels
        t = min(self.decode_step, iters)
        self.opt.zero_grad(set_to_none=True)
        noised = img + torch.randn_like(img) * self.noise_std
        all_levels = self.model(noised, iters=iters, return_all=True)
        top = all_levels[t, :, :, -1]
        recon = self.decoder(top)
        loss = F.mse_loss(recon.float(), img.float())
        loss.backward()
        if self.distributed:
            self.ddp_model.finalize()
            self.ddp_dec.finalize()
            self.heartbeat.tick()
        if self.master is not None:
            with torch.no_grad():
                for mw, q in zip(self.master, self._params):
                    # None (not stale) when the param got no grad this
                    # step, e.g. init_levels on the stateful path
                    mw.grad = q.grad.float() if q.grad is not None else None
                if self.grad_clip:
                    torch.nn.utils.clip_grad_norm_(self.master,
                                                   self.grad_clip)
                self.opt.step()
                for mw, q in zip(self.master, self._params):
                    q.data.copy_(mw)
        else:
            if self.grad_clip:
                torch.nn.utils.clip_grad_norm_(
                    [q for g in self.opt.param_groups for q in g["params"]],
                    self.grad_clip)
            self.opt.step()
        self.step_idx += 1
        return loss.item() if sync_loss else loss.detach()

    def log(self, **metrics):
        if self.log_path:
            with open(self.log_path, "a") as f:
                f.write(json.dumps({"step": self.step_idx,
                                    "time": time.time(), **metrics}) + "\n")

    # -------------------- checkpoint / resume --------------------

    def save_checkpoint(self, path: str):
        if self.distributed and dist.get_rank() != 0:
            return
        tmp = path + ".tmp"
        torch.save({
            "model": self.model.state_dict(),
            "decoder": self.decoder.state_dict(),
            "optimizer": self.opt.state_dict(),
            "master": self.master,
            "step": self.step_idx,
            "rng": torch.get_rng_state(),
            "cuda_rng": (torch.cuda.get_rng_state()
                         if torch.cuda.is_available() else None),
        }, tmp)
        os.replace(tmp, path)

    def load_checkpoint(self, path: str):
        ckpt = torch.load(path, map_location="cpu", weights_only=False)
        self.model.load_state_dict(ckpt["model"])
        self.decoder.load_state_dict(ckpt["decoder"])
        self.opt.load_state_dict(ckpt["optimizer"])
        if ckpt.get("master") is not None and self.master is not None:
            for mw, saved in zip(self.master, ckpt["master"]):
                mw.data.copy_(saved.to(mw.device))
        self.step_idx = ckpt["step"]
        torch.set_rng_state(ckpt["rng"])
        if ckpt.get("cuda_rng") is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(ckpt["cuda_rng"])
        return self
