"""Denoising self-supervised trainer for GLOM (the reference's training
recipe, README.md:56-90): noise the image, run the model with return_all,
decode the top level at a mid-iteration timestep back to pixels, MSE against
the clean image. Adds what the reference leaves to the user: optimizer,
data-parallel gradient sync, checkpointing, metrics — and, on the native
GPU path, hipGraph capture of the ENTIRE training step (noise + forward +
backward + fused AdamW), replayed as a single graph launch per step."""

from __future__ import annotations

import json
import os
import time
import warnings

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
                 process_group=None, graph_step=True, overlap_tail=True,
                 micro_batches=1, log_path=None):
        """``process_group``: the group whose ranks are data-parallel
        replicas (default: the global group). For 2-D DP x SP meshes pass
        the DP subgroup here — BucketedDDP AVERAGES over its group, which
        is correct for DP replicas; SP shards need a separate SUM
        all-reduce over the SP subgroup (see parallel/sequence.py).

        ``graph_step``: hipGraph-capture the whole training step on the
        native bf16 GPU path (keyed on batch shape + iters) and replay it;
        falls back to eager launches automatically if capture fails.
        Disable with graph_step=False or GLOM_NO_GRAPH_STEP=1."""
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
        self._params = params
        # bf16 training keeps fp32 master weights: updates of size lr*grad
        # would otherwise partially round away in bf16 parameter storage
        self.master = None
        self.fused_opt = None
        if p.dtype == torch.bfloat16:
            self.master = [q.detach().float().requires_grad_(False)
                           for q in params]
            if p.is_cuda and self._ops_available():
                # single-kernel AdamW: bf16 grads -> fp32 m/v/master ->
                # bf16 params, with the global-norm clip fused
                from glom_pytorch_amd.ops.optim import FusedAdamW
                self.fused_opt = FusedAdamW(
                    params, self.master, lr=lr, weight_decay=1e-2,
                    max_grad_norm=grad_clip if grad_clip else 0.0)
                self.opt = self.fused_opt
            else:
                self.opt = torch.optim.AdamW(self.master, lr=lr,
                                             foreach=True)
        else:
            self.opt = torch.optim.AdamW(params, lr=lr, foreach=True)
        self.grad_clip = grad_clip
        self.step_idx = 0
        self.log_path = log_path
        self.graph_step = (graph_step
                           and os.environ.get("GLOM_NO_GRAPH_STEP", "0")
                           != "1")
        self.overlap_tail = (overlap_tail
                             and os.environ.get("GLOM_NO_OVERLAP_TAIL",
                                                "0") != "1")
        # pipelined microbatching: split the batch in `micro_batches`
        # chunks; each chunk's loss is scaled 1/m so the summed grads
        # equal the full-batch gradient EXACTLY (mean-MSE is linear in
        # batch averaging); chunk i+1's forward runs on a side stream
        # overlapping chunk i's backward.
        self.micro_batches = int(os.environ.get("GLOM_MICRO_BATCHES",
                                                str(micro_batches)))
        self._micro_stream = None
        self._graphs: dict = {}

    @staticmethod
    def _ops_available():
        from glom_pytorch_amd import ops
        return ops.available()

    # ------------------------------ stepping -------------------------- #

    def step(self, img: torch.Tensor, iters: int | None = None,
             sync_loss: bool = True):
        """One denoising training step. With sync_loss=False the loss is
        returned as a device tensor (no host sync), letting the host run
        ahead and queue the next step's launches."""
        iters = iters if iters is not None else 2 * self.model.levels
        if self._graph_usable(img):
            loss = self._graphed_step(img, iters)
        else:
            loss = self._eager_step(img, iters)
        self.step_idx += 1
        if self.heartbeat is not None:
            self.heartbeat.tick()
        return loss.item() if sync_loss else loss.detach()

    def _graph_usable(self, img) -> bool:
        # micro-pipelining accumulates grads across streams; the
        # AccumulateGrad node's stream mismatch breaks hipGraph capture
        # (torch warns, replay faults) — micro mode runs eager-launched
        return (self.graph_step and self.micro_batches == 1
                and self.fused_opt is not None
                and img.is_cuda and img.dtype == torch.bfloat16
                and not getattr(self.model, "force_eager", False)
                and os.environ.get("GLOM_FORCE_EAGER", "0") != "1")

    def _eager_step(self, img: torch.Tensor, iters: int) -> torch.Tensor:
        m = self.micro_batches
        if (m > 1 and img.is_cuda and img.shape[0] % m == 0
                and not getattr(self.model, "force_eager", False)):
            return self._micro_step(img, iters, m)
        t = min(self.decode_step, iters)
        self.opt.zero_grad(set_to_none=True)
        noised = img + torch.randn_like(img) * self.noise_std
        # the loss reads only trajectory time t: iterations >= t carry
        # exactly-zero gradient, so they run forward-only (grad_iters) —
        # identical values and gradients, ~(iters-t)/iters less backward.
        # overlap_tail additionally runs them on a side stream so they
        # overlap the backward below; join_tail_stream() before the
        # optimizer guarantees no weight write races the tail's reads.
        overlap = (self.overlap_tail and img.is_cuda
                   and not getattr(self.model, "force_eager", False))
        all_levels = self.model(noised, iters=iters, return_all=True,
                                grad_iters=t, overlap_tail=overlap)
        top = all_levels[t, :, :, -1]
        recon = self.decoder(top)
        loss = F.mse_loss(recon.float(), img.float())
        loss.backward()
        if self.distributed:
            self.ddp_model.finalize()
            self.ddp_dec.finalize()
        if overlap:
            from glom_pytorch_amd.ops.functional import join_tail_stream
            join_tail_stream()
        if self.fused_opt is not None:
            self.fused_opt.step()
        elif self.master is not None:
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
        return loss.detach()

    def _micro_forward(self, half: torch.Tensor, iters: int, t: int,
                       scale: float) -> torch.Tensor:
        noised = half + torch.randn_like(half) * self.noise_std
        traj = self.model(noised, iters=iters, return_all=True,
                          grad_iters=t, overlap_tail=True)
        recon = self.decoder(traj[t, :, :, -1])
        return F.mse_loss(recon.float(), half.float()) * scale

    def _micro_step(self, img: torch.Tensor, iters: int,
                    m: int) -> torch.Tensor:
        """Pipelined microbatches: forward of chunk i+1 (on a side
        stream) overlaps backward of chunk i. Gradients are EXACTLY the
        full-batch gradient: mean-MSE over the batch = (1/m) sum of the
        chunk means, and autograd sums the scaled chunk grads."""
        t = min(self.decode_step, iters)
        self.opt.zero_grad(set_to_none=True)
        if self._micro_stream is None:
            self._micro_stream = torch.cuda.Stream()
        s_f = self._micro_stream
        halves = img.chunk(m)
        if self.distributed:
            self.ddp_model.set_accumulate(True)
            self.ddp_dec.set_accumulate(True)
        cur = torch.cuda.current_stream()
        losses = []
        pending = self._micro_forward(halves[0], iters, t, 1.0 / m)
        for i in range(1, m):
            # fork chunk i's forward before launching chunk i-1's backward
            ev = torch.cuda.Event()
            ev.record(cur)
            with torch.cuda.stream(s_f):
                s_f.wait_event(ev)
                nxt = self._micro_forward(halves[i], iters, t, 1.0 / m)
            pending.backward()
            losses.append(pending.detach())
            if self.distributed and i == m - 1:
                self.ddp_model.set_accumulate(False)
                self.ddp_dec.set_accumulate(False)
                # counts were consumed? hooks were suppressed: reset
                self.ddp_model._reset_counts()
                self.ddp_dec._reset_counts()
            pending = nxt
        pending.backward()   # engine runs this chunk's ops on s_f
        losses.append(pending.detach())
        cur.wait_stream(s_f)
        if self.distributed:
            self.ddp_model.finalize()
            self.ddp_dec.finalize()
        from glom_pytorch_amd.ops.functional import join_tail_stream
        join_tail_stream()
        loss = losses[0]
        for l in losses[1:]:
            loss = loss + l
        if self.fused_opt is not None:
            self.fused_opt.step()
        elif self.master is not None:
            with torch.no_grad():
                for mw, q in zip(self.master, self._params):
                    mw.grad = (q.grad.float() if q.grad is not None
                               else None)
                if self.grad_clip:
                    torch.nn.utils.clip_grad_norm_(self.master,
                                                   self.grad_clip)
                self.opt.step()
                for mw, q in zip(self.master, self._params):
                    q.data.copy_(mw)
        else:
            self.opt.step()
        return loss.detach()

    # --------------------- hipGraph-captured step --------------------- #

    def _graphed_step(self, img: torch.Tensor, iters: int) -> torch.Tensor:
        key = (tuple(img.shape), iters)
        entry = self._graphs.get(key)
        if entry is None:
            entry = self._capture(img, iters)
            self._graphs[key] = entry
        if entry is False:
            return self._eager_step(img, iters)
        entry["img"].copy_(img)
        entry["graph"].replay()
        self.fused_opt.note_replays(1)
        return entry["loss"]

    def _capture(self, img: torch.Tensor, iters: int):
        """Record noise + forward (all `iters` GLOM steps) + backward +
        bucketed all-reduce (if distributed) + fused AdamW as ONE hipGraph.
        torch.cuda.graph handles the philox RNG state, so every replay
        draws fresh noise. Any capture failure falls back to eager."""
        if self.distributed and os.environ.get("GLOM_GRAPH_DIST",
                                               "1") == "0":
            return False
        try:
            static_img = img.clone()
            # snapshot optimizer/weight state: the warmup steps below are
            # real updates, rolled back so a graphed step == one step
            fo = self.fused_opt
            with torch.no_grad():
                snap = ([m.clone() for m in self.master]
                        + [t.clone() for t in fo.exp_avg]
                        + [t.clone() for t in fo.exp_avg_sq]
                        + [q.detach().clone() for q in self._params]
                        + [fo._step_dev.clone()])
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):     # allocator/runtime warmup
                    self._eager_step(static_img, iters)
            torch.cuda.current_stream().wait_stream(s)
            with torch.no_grad():
                live = (self.master + fo.exp_avg + fo.exp_avg_sq
                        + [q.data for q in self._params] + [fo._step_dev])
                for dst, src in zip(live, snap):
                    dst.copy_(src)
                fo._step_host = max(0, fo._step_host - 2)
            # start from grads=None: backward inside capture then ASSIGNS
            # fresh graph-pool grad tensors (no accumulate node recorded)
            for q in self._params:
                q.grad = None
            graph = torch.cuda.CUDAGraph()
            # (a) flush pending destructors NOW and keep the cyclic GC off
            # during capture: a stale CUDAGraph/tensor destructor firing
            # mid-capture makes HIP API calls that invalidate the capture
            # (the replay then segfaults); (b) thread_local error mode:
            # the RCCL watchdog thread polls events on this device during
            # capture and must not invalidate it either.
            import gc
            gc.collect()
            gc_was_enabled = gc.isenabled()
            gc.disable()
            try:
                with torch.cuda.graph(graph,
                                      capture_error_mode="thread_local"):
                    loss = self._eager_step(static_img, iters)
            finally:
                if gc_was_enabled:
                    gc.enable()
            # capture RECORDS the step without executing it: undo the
            # Python-side count bump from the captured call
            fo._step_host = max(0, fo._step_host - 1)
            return {"graph": graph, "img": static_img, "loss": loss}
        except Exception as e:  # pragma: no cover - GPU/runtime specific
            warnings.warn(f"training-step hipGraph capture failed "
                          f"({type(e).__name__}: {e}); using eager "
                          f"launches")
            return False

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
        # params follow the restored masters (the fused path reads/writes
        # masters as the source of truth)
        if self.master is not None:
            with torch.no_grad():
                for mw, q in zip(self.master, self._params):
                    q.data.copy_(mw.to(q.dtype))
        torch.set_rng_state(ckpt["rng"])
        if ckpt.get("cuda_rng") is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(ckpt["cuda_rng"])
        # drop any captured graphs: they bake in old grad/param pointers
        self._graphs.clear()
        return self
