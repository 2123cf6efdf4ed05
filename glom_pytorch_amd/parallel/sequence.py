"""Sequence parallelism: shard the N patch columns across ranks.

The reference has no parallelism at all ("Parallelism
strategies: NONE"); DP over images is this framework's mandated strategy
(`parallel/ddp.py`). This module adds the §5 "later extension": GLOM's only
cross-column op is consensus attention (reference glom_pytorch.py:56-73), so
the patch dimension shards cleanly — bottom-up / top-down MLPs and the level
mix are purely column-local, and only attention communicates.

Two attention exchange modes:

- ``mode="allgather"`` (default, differentiable): each rank all-gathers the
  full ``levels`` tensor once per iteration (via
  ``torch.distributed.nn.functional.all_gather``, so gradients flow back and
  SP *training* works; composing with the DP trainer for a 2-D DP x SP mesh
  additionally needs the DP subgroup passed as ``DenoisingTrainer``'s
  ``process_group`` — BucketedDDP averages over its group, while SP shards
  of one replica need their weight grads SUM-reduced over the SP subgroup,
  as tests/test_sequence_parallel_cpu.py does), then computes attention for
  its local query rows only. One bucketed collective per iteration —
  the RCCL/xGMI-friendly shape: ring all-gather over 7 point-to-point links,
  each rank receives (W-1)/W of B·N·L·d once.

- ``mode="ring"`` (inference): k̂/v blocks rotate around the ring
  (``isend``/``irecv`` point-to-point — the native xGMI pattern), queries
  stay local, and the softmax is accumulated with the standard online-max
  trick. The reference's two masks are applied at *global* column indices
  per block: the −5e-4 self value on the diagonal (NOT −inf — reference
  glom_pytorch.py:11) and the −dtype.max local-radius mask. Exactly equal
  to the dense softmax up to summation order.

Both modes reproduce ``Glom._eager_forward`` semantics exactly; parity is
pinned by tests/test_sequence_parallel_cpu.py (gloo, world_size 2) on CPU —
on GPU the same code runs over RCCL.

N must be divisible by world_size (N=256 and N=1024 shard evenly at any
world size up to 8).
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn.functional as F
from einops import rearrange, repeat

from glom_pytorch_amd.models.glom import TOKEN_ATTEND_SELF_VALUE


def _shard_bounds(n: int, pg=None):
    ws = dist.get_world_size(pg)
    rank = dist.get_rank(pg)
    if n % ws != 0:
        raise ValueError(f"num_patches {n} not divisible by world size {ws}")
    per = n // ws
    return rank * per, (rank + 1) * per, rank, ws


class _ContigGrad(torch.autograd.Function):
    """Force incoming gradients contiguous.

    ``torch.distributed.nn.functional.all_gather``'s backward feeds the
    upstream grads straight into an all-to-all, which silently computes
    garbage for non-contiguous tensors (einsum backwards routinely produce
    permuted-view grads). This shim sits between the gather and its
    consumers so the grads entering the collective are always contiguous.
    """

    @staticmethod
    def forward(ctx, x):
        return x.view_as(x)

    @staticmethod
    def backward(ctx, g):
        return g.contiguous()


def _full_levels(levels_local: torch.Tensor, pg=None) -> torch.Tensor:
    """Differentiable all-gather of the column shards -> (B, N, L, d)."""
    if torch.is_grad_enabled() and levels_local.requires_grad:
        from torch.distributed.nn.functional import all_gather
        parts = [_ContigGrad.apply(t)
                 for t in all_gather(levels_local.contiguous(), group=pg)]
    else:
        ws = dist.get_world_size(pg)
        parts = [torch.empty_like(levels_local) for _ in range(ws)]
        dist.all_gather(parts, levels_local.contiguous(), group=pg)
    return torch.cat(parts, dim=1)


def _attention_allgather(model, levels_local, n0, n1, pg):
    """Local-row consensus against the gathered full column set."""
    att = model.attention
    full = _full_levels(levels_local, pg)
    d = full.shape[-1]
    q = levels_local
    k = F.normalize(full, dim=-1)

    sim = torch.einsum("b i l d, b j l d -> b l i j", q, k) * (d ** -0.5)

    if not att.attend_self:
        n = full.shape[1]
        gi = torch.arange(n0, n1, device=sim.device)
        gj = torch.arange(n, device=sim.device)
        eye = gi[:, None] == gj[None, :]
        sim = sim.masked_fill(eye.view(1, 1, n1 - n0, n),
                              TOKEN_ATTEND_SELF_VALUE)
    if att.local_consensus_radius > 0:
        sim = sim.masked_fill(att.non_local_mask[:, n0:n1].unsqueeze(1),
                              -torch.finfo(sim.dtype).max)

    attn = sim.softmax(dim=-1)
    return torch.einsum("b l i j, b j l d -> b i l d", attn, full)


def _ring_rotate(block: torch.Tensor, rank: int, ws: int, pg=None):
    """One ring step: send my block to rank+1, receive rank-1's."""
    nxt, prv = (rank + 1) % ws, (rank - 1) % ws
    recv = torch.empty_like(block)
    ops = [dist.P2POp(dist.isend, block.contiguous(), nxt, group=pg),
           dist.P2POp(dist.irecv, recv, prv, group=pg)]
    for req in dist.batch_isend_irecv(ops):
        req.wait()
    return recv


@torch.no_grad()
def _attention_ring(model, levels_local, n0, n1, num_patches, pg):
    """Blockwise consensus: k̂/v rotate around the ring, online softmax.

    Accumulators (fp32): running max m, running sum s (both per (b,l,i)),
    running unnormalized output o. Each arriving block carries the columns
    of rank (rank - step) % ws; masks use those global indices.
    """
    att = model.attention
    b, nloc, L, d = levels_local.shape
    _, _, rank, ws = _shard_bounds(num_patches, pg)
    scale = d ** -0.5
    dt = levels_local.dtype

    q = levels_local
    khat = F.normalize(levels_local, dim=-1)
    # one rotating buffer: [0]=k̂ block, [1]=v block
    blk = torch.stack((khat, levels_local)).contiguous()

    m = torch.full((b, L, nloc, 1), -torch.inf, device=q.device)
    s = torch.zeros(b, L, nloc, 1, device=q.device)
    o = torch.zeros(b, nloc, L, d, device=q.device)
    gi = torch.arange(n0, n1, device=q.device)
    neg = -torch.finfo(dt).max

    for step in range(ws):
        src = (rank - step) % ws
        j0 = src * nloc
        sim = torch.einsum("b i l d, b j l d -> b l i j",
                           q, blk[0]).float() * scale

        if not att.attend_self:
            gj = torch.arange(j0, j0 + nloc, device=q.device)
            eye = gi[:, None] == gj[None, :]
            sim = sim.masked_fill(eye.view(1, 1, nloc, nloc),
                                  float(torch.tensor(TOKEN_ATTEND_SELF_VALUE,
                                                     dtype=dt)))
        masked = None
        if att.local_consensus_radius > 0:
            masked = att.non_local_mask[:, n0:n1, j0:j0 + nloc].unsqueeze(1)
            sim = sim.masked_fill(masked, neg)

        bm = sim.amax(dim=-1, keepdim=True)
        m_new = torch.maximum(m, bm)
        corr = torch.exp(m - m_new)
        p = torch.exp(sim - m_new)
        if masked is not None:
            # a fully-masked block has bm == neg; exp(neg - neg) == 1 would
            # leak mass, so masked entries are zeroed explicitly
            p = p.masked_fill(masked, 0.0)
        s = s * corr + p.sum(dim=-1, keepdim=True)
        o = o * rearrange(corr, "b l i () -> b i l ()") + torch.einsum(
            "b l i j, b j l d -> b i l d", p, blk[1].float())
        m = m_new
        if step + 1 < ws:
            blk = _ring_rotate(blk, rank, ws, pg)

    return (o / rearrange(s, "b l i () -> b i l ()")).to(dt)


def sp_forward(model, img, iters: int | None = None,
               levels: torch.Tensor | None = None,
               return_all: bool = False, mode: str = "allgather",
               process_group=None, gather_output: bool = False):
    """Sequence-parallel ``Glom.forward``: every rank gets the SAME image
    batch and computes the columns [n0, n1) of the patch grid.

    Returns the local column shard (B, N/ws, L, d) — or the full (B, N, L,
    d) when ``gather_output=True``. ``levels`` (stateful continuation) may
    be the full tensor or the local shard. ``mode="ring"`` requires
    no_grad; ``mode="allgather"`` is differentiable (compose with the DP
    trainer for 2-D DP×SP meshes).
    """
    if mode not in ("allgather", "ring"):
        raise ValueError(f"unknown sp mode {mode!r}")
    if mode == "ring" and torch.is_grad_enabled():
        raise RuntimeError("ring mode is inference-only; wrap in no_grad() "
                           "or use mode='allgather' for training")
    iters = iters if iters is not None else 2 * model.levels
    pg = process_group

    tokens = model.image_to_tokens(img)
    n = tokens.shape[1]
    n0, n1, _, _ = _shard_bounds(n, pg)

    bottom = rearrange(tokens[:, n0:n1], "b n d -> b n () d")
    pos = model.pos_emb(torch.arange(n0, n1, device=img.device))
    pos = rearrange(pos, "n d -> () n () d")

    if levels is None:
        lv = repeat(model.init_levels, "l d -> b n l d",
                    b=img.shape[0], n=n1 - n0)
    elif levels.shape[1] == n:
        lv = levels[:, n0:n1]
    elif levels.shape[1] == n1 - n0:
        lv = levels
    else:
        raise ValueError(f"levels has {levels.shape[1]} columns; expected "
                         f"{n} (full) or {n1 - n0} (local shard)")

    # bf16-on-GPU ranks run their local column shard through the CDNA4
    # grouped-GEMM / level-mix kernels (the FF work, ~3/4 of the FLOPs, is
    # column-local); only the attention exchange stays in torch ops.
    use_native = model._use_native(img)
    if use_native:
        from glom_pytorch_amd.ops.functional import GroupedFFFn, LevelMixFn
        tokens_loc = tokens[:, n0:n1].contiguous()
        pos_loc = model.pos_emb.weight[n0:n1].contiguous()
        bw, tw = model.bottom_up.net, model.top_down.net
        lv = lv.contiguous()

    contrib = model._contrib.to(lv.dtype)
    trajectory = [lv]
    for _ in range(iters):
        if mode == "allgather":
            consensus = _attention_allgather(model, lv, n0, n1, pg)
        else:
            consensus = _attention_ring(model, lv, n0, n1, n, pg)
        if use_native:
            bu = GroupedFFFn.apply(tokens_loc, lv, None,
                                   bw[1].weight[..., 0], bw[1].bias,
                                   bw[3].weight[..., 0], bw[3].bias, 0)
            td = GroupedFFFn.apply(None, lv, pos_loc,
                                   tw[1].weight[..., 0], tw[1].bias,
                                   tw[3].weight[..., 0], tw[3].bias, 1)
            lv = LevelMixFn.apply(lv, bu, td, consensus.contiguous())
        else:
            bu_in = torch.cat((bottom, lv[..., :-1, :]), dim=-2)
            bu = model.bottom_up(bu_in)
            td = model.top_down(lv[..., 1:, :] + pos)
            td = F.pad(td, (0, 0, 0, 1), value=0.0)
            lv = (lv + bu + td + consensus) / contrib
        trajectory.append(lv)

    out = torch.stack(trajectory) if return_all else lv
    if gather_output:
        dim = 2 if return_all else 1
        if torch.is_grad_enabled() and out.requires_grad:
            from torch.distributed.nn.functional import all_gather
            parts = [_ContigGrad.apply(t)
                     for t in all_gather(out.contiguous(), group=pg)]
        else:
            parts = [torch.empty_like(out)
                     for _ in range(dist.get_world_size(pg))]
            dist.all_gather(parts, out.contiguous(), group=pg)
        out = torch.cat(parts, dim=dim)
    return out
