"""The GLOM model (Hinton, arXiv:2102.12627) with the reference public API.

API / state_dict parity contract (checked by tests/test_state_dict.py):
  - ``Glom(dim, levels, image_size, patch_size, consensus_self,
    local_consensus_radius)`` — reference glom_pytorch.py:78-108
  - ``forward(img, iters=None, levels=None, return_all=False)`` — reference
    glom_pytorch.py:110-150
  - state_dict keys/shapes: init_levels, image_to_tokens.1.{weight,bias},
    pos_emb.weight, bottom_up.net.{1,3}.{weight,bias},
    top_down.net.{1,3}.{weight,bias}, attention.non_local_mask (iff
    local_consensus_radius > 0) — reference glom_pytorch.py:94-108,23-36,54

Compute dispatch (this file is the thin PyTorch-ROCm layer; all heavy lifting
is below it):
  - CPU / fp32, or ``GLOM_FORCE_EAGER=1``: a plain-PyTorch eager path.
  - bf16 tensors on a gfx950 GPU: hand-written CDNA4 HIP kernels via
    ``glom_pytorch_amd.ops`` (grouped MFMA GEMM family, fully fused
    consensus attention incl. masked softmax and AV, fused level mixing,
    hand-written backward for everything). This path raises if the HIP
    extension is not importable on a GPU machine — there is no silent
    eager fallback on GPU.
"""

from __future__ import annotations

import math
import os

import torch
import torch.nn.functional as F
from torch import nn
from einops import rearrange, repeat
from einops.layers.torch import Rearrange

# The mild (NOT -inf) penalty the reference writes onto the self-attention
# diagonal when consensus_self=False (reference glom_pytorch.py:11).
TOKEN_ATTEND_SELF_VALUE = -5e-4


class GroupedFeedForward(nn.Module):
    """L independent per-level MLPs (d -> mult*d -> GELU -> d).

    Stored as grouped 1x1 Conv1d over channel layout ``b (l d) n`` purely for
    state_dict parity with the reference (glom_pytorch.py:23-36): weight
    shapes are ``(groups*dim*mult, dim, 1)`` and ``(groups*dim, dim*mult, 1)``.
    The HIP engine consumes these weights directly as G packed GEMM operands.
    """

    def __init__(self, *, dim: int, groups: int, mult: int = 4):
        super().__init__()
        self.dim = dim
        self.groups = groups
        self.mult = mult
        total = dim * groups
        self.net = nn.Sequential(
            Rearrange("b n l d -> b (l d) n"),
            nn.Conv1d(total, total * mult, 1, groups=groups),
            nn.GELU(),
            nn.Conv1d(total * mult, total, 1, groups=groups),
            Rearrange("b (l d) n -> b n l d", l=groups),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


class ConsensusAttention(nn.Module):
    """Per-level spatial attention across the N patch columns.

    q = levels (raw), k = L2-normalized levels, v = levels (raw); scale is
    d^-0.5 applied to the raw dot products (reference glom_pytorch.py:38-73).
    Optional self-mask (diagonal := -5e-4) and optional local-radius mask
    (non-local := -dtype.max) are applied to the scaled similarities before
    the softmax over source columns j.
    """

    def __init__(self, num_patches_side: int, attend_self: bool = True,
                 local_consensus_radius: int = 0):
        super().__init__()
        self.attend_self = attend_self
        self.local_consensus_radius = local_consensus_radius

        if local_consensus_radius > 0:
            # 2-D patch-grid coordinates -> pairwise euclidean distance ->
            # boolean "too far to attend" mask, as a persistent buffer so the
            # state_dict matches the reference (glom_pytorch.py:45-54).
            side = num_patches_side
            hh, ww = torch.meshgrid(torch.arange(side), torch.arange(side),
                                    indexing="ij")
            coords = torch.stack((hh, ww)).float()
            coords = rearrange(coords, "c h w -> (h w) c")
            pairwise = torch.cdist(coords, coords)
            mask = pairwise > float(local_consensus_radius)
            self.register_buffer("non_local_mask", rearrange(mask, "i j -> () i j"))

    def forward(self, levels: torch.Tensor) -> torch.Tensor:
        n, d = levels.shape[1], levels.shape[-1]
        q = levels
        k = F.normalize(levels, dim=-1)

        sim = torch.einsum("b i l d, b j l d -> b l i j", q, k) * (d ** -0.5)

        if not self.attend_self:
            eye = torch.eye(n, device=levels.device, dtype=torch.bool)
            sim = sim.masked_fill(eye.view(1, 1, n, n), TOKEN_ATTEND_SELF_VALUE)

        if self.local_consensus_radius > 0:
            sim = sim.masked_fill(self.non_local_mask,
                                  -torch.finfo(sim.dtype).max)

        attn = sim.softmax(dim=-1)
        return torch.einsum("b l i j, b j l d -> b i l d", attn, levels)


class Glom(nn.Module):
    """GLOM: iterative column/level refinement over an image patch grid.

    Each of the N = (image_size/patch_size)^2 patches owns a column of
    ``levels`` embeddings of width ``dim``; for ``iters`` timesteps every
    level is replaced by the average of {previous value, bottom-up MLP of the
    level below (patch tokens act as level -1), top-down MLP of the level
    above (+ positional embedding), consensus attention across columns}.
    The top level has no top-down input and averages 3 contributions
    (reference glom_pytorch.py:110-150).
    """

    def __init__(self, *, dim: int = 512, levels: int = 6,
                 image_size: int = 224, patch_size: int = 14,
                 consensus_self: bool = False,
                 local_consensus_radius: int = 0):
        super().__init__()
        if levels < 2:
            raise ValueError("Glom needs levels >= 2 (top_down has levels-1 groups)")
        side = image_size // patch_size
        self.levels = levels
        self.dim = dim
        self.image_size = image_size
        self.patch_size = patch_size
        self.num_patches_side = side
        self.num_patches = side * side

        self.image_to_tokens = nn.Sequential(
            Rearrange("b c (h p1) (w p2) -> b (h w) (p1 p2 c)",
                      p1=patch_size, p2=patch_size),
            nn.Linear(patch_size ** 2 * 3, dim),
        )
        self.pos_emb = nn.Embedding(self.num_patches, dim)
        self.init_levels = nn.Parameter(torch.randn(levels, dim))

        self.bottom_up = GroupedFeedForward(dim=dim, groups=levels)
        self.top_down = GroupedFeedForward(dim=dim, groups=levels - 1)

        self.attention = ConsensusAttention(
            side, attend_self=consensus_self,
            local_consensus_radius=local_consensus_radius)

        # num_contributions per level: every level averages 4 contributions
        # except the top, which has no top-down input and averages 3
        # (reference glom_pytorch.py:128-129,141-144).
        contrib = torch.full((levels,), 4.0)
        contrib[-1] = 3.0
        self.register_buffer("_contrib", contrib.view(1, 1, levels, 1),
                             persistent=False)

    # ------------------------------------------------------------------ #

    def _native_capable(self) -> bool:
        """Configs the HIP engine covers: dim a multiple of 8 (16B bf16
        vector loads) and <= 16 levels (the grouped-GEMM pointer-table
        width). Everything the reference README exercises qualifies;
        exotic configs run the eager path with a one-time warning."""
        return self.dim % 8 == 0 and self.levels <= 16

    def _use_native(self, img: torch.Tensor) -> bool:
        if not img.is_cuda:
            return False
        if os.environ.get("GLOM_FORCE_EAGER", "0") == "1":
            return False
        if getattr(self, "force_eager", False):
            return False
        if img.dtype != torch.bfloat16:
            return False
        if not self._native_capable():
            if not getattr(self, "_warned_capability", False):
                import warnings
                warnings.warn(
                    f"Glom(dim={self.dim}, levels={self.levels}) is outside "
                    "the HIP engine's envelope (dim%8==0, levels<=16); "
                    "running the eager path on GPU.")
                self._warned_capability = True
            return False
        return True

    def forward(self, img: torch.Tensor, iters: int | None = None,
                levels: torch.Tensor | None = None,
                return_all: bool = False,
                grad_iters: int | None = None,
                overlap_tail: bool = False) -> torch.Tensor:
        """grad_iters (extension kwarg, default None = exact reference
        autograd graph): run iterations >= grad_iters under no_grad.
        Forward values are unchanged; use when the loss only reads
        trajectory times <= grad_iters (e.g. the denoising recipe) to skip
        backprop through post-loss iterations whose gradient contribution
        is exactly zero. ``overlap_tail`` (native path only) additionally
        runs those iterations on a side stream so they overlap the
        caller's backward; the caller MUST then call
        ``ops.functional.join_tail_stream()`` before reading the tail's
        trajectory slices or mutating weights."""
        if img.dim() != 4 or img.shape[1] != 3:
            raise ValueError(f"expected image batch (b, 3, H, W), got {tuple(img.shape)}")
        iters = iters if iters is not None else 2 * self.levels

        if self._use_native(img):
            cache = getattr(self, "_graph_cache", None)
            if cache is not None:
                from glom_pytorch_amd.graphs import graphs_usable
                if graphs_usable(self, img, levels):
                    return cache.run(img, iters, levels, return_all)
            from glom_pytorch_amd.ops import native_forward
            return native_forward(self, img, iters=iters, levels=levels,
                                  return_all=return_all,
                                  grad_iters=grad_iters,
                                  overlap_tail=overlap_tail)
        return self._eager_forward(img, iters, levels, return_all,
                                   grad_iters)

    def enable_graphs(self):
        """hipGraph-capture the T-step loop for inference: each distinct
        (shape, iters, return_all, stateful) forward is recorded once and
        replayed as a single graph launch. No-grad forwards only; training
        and CPU paths are unaffected."""
        from glom_pytorch_amd.graphs import GraphCache
        self._graph_cache = GraphCache(self)
        return self

    def disable_graphs(self):
        self._graph_cache = None
        return self

    # ------------------------------------------------------------------ #
    # Eager (plain PyTorch) path — the semantic specification. CPU tests
    # compare this against an independent loop-level oracle, and GPU tests
    # compare the HIP engine against it.

    def _eager_forward(self, img, iters, levels, return_all,
                       grad_iters=None):
        b = img.shape[0]
        tokens = self.image_to_tokens(img)
        n = tokens.shape[1]

        pos = self.pos_emb(torch.arange(n, device=img.device))
        pos = rearrange(pos, "n d -> () n () d")

        bottom = rearrange(tokens, "b n d -> b n () d")
        if levels is None:
            levels = repeat(self.init_levels, "l d -> b n l d", b=b, n=n)

        trajectory = [levels]
        for t in range(iters):
            if grad_iters is not None and t >= grad_iters:
                with torch.no_grad():
                    levels = self._eager_step(bottom, levels, pos)
            else:
                levels = self._eager_step(bottom, levels, pos)
            trajectory.append(levels)

        if return_all:
            return torch.stack(trajectory)
        return levels

    def _eager_step(self, bottom, levels, pos):
        # bottom-up sees [input, level_0 .. level_{L-2}] and predicts levels
        # 0..L-1; top-down sees [level_1 .. level_{L-1}] + pos and predicts
        # levels 0..L-2 (zero for the top slot).
        bu_in = torch.cat((bottom, levels[..., :-1, :]), dim=-2)
        bu = self.bottom_up(bu_in)

        td_in = levels[..., 1:, :] + pos
        td = self.top_down(td_in)
        td = F.pad(td, (0, 0, 0, 1), value=0.0)

        consensus = self.attention(levels)

        return (levels + bu + td + consensus) / self._contrib.to(levels.dtype)
