"""Autograd wrappers around the CDNA4 HIP ops + the native Glom forward.

Every hot op of the reference forward is a custom
torch.autograd.Function with hand-written HIP forward AND backward kernels;
autograd only stitches the iteration loop together (and re-traverses it for
losses attached at any (time, level) of the trajectory, as in the
reference's denoising recipe, README.md:56-90).
"""

from __future__ import annotations

import os

import torch

from glom_pytorch_amd.ops import _load_extension


class PatchEmbedFn(torch.autograd.Function):
    """Patchify + embed GEMM (K1; reference glom_pytorch.py:94-97,114):
    the rearrange runs as a HIP gather kernel into a K-padded (B,N,Kp)
    token matrix and the 588->dim projection on the tuned NT MFMA kernel.
    Backward: split-K TN weight grad, native colsum bias grad, and (only
    when the image itself needs grad) an NN GEMM + scatter back to
    (B,3,H,W)."""

    @staticmethod
    def forward(ctx, img, w, b, patch_size):
        ext = _load_extension()
        tokens, X = ext.patch_embed_fwd(img.contiguous(), w, b, patch_size)
        ctx.save_for_backward(X, w)
        ctx.patch = patch_size
        ctx.img_hw = (img.shape[2], img.shape[3])
        return tokens

    @staticmethod
    def backward(ctx, dTokens):
        ext = _load_extension()
        X, w = ctx.saved_tensors
        need_dimg = ctx.needs_input_grad[0]
        dW, dB, dImg = ext.patch_embed_bwd(
            dTokens.contiguous(), X, w, ctx.patch, ctx.img_hw[0],
            ctx.img_hw[1], need_dimg)
        return (dImg if need_dimg else None), dW, dB, None


class TrajectoryFn(torch.autograd.Function):
    """return_all trajectory (reference glom_pytorch.py:126,145-148)
    without the torch.stack copy: every step already wrote its output into
    slab[t+1] (and slab[0] holds the initial state), so forward is an
    identity on the slab and backward just slices the incoming gradient
    back to the per-step outputs."""

    @staticmethod
    def forward(ctx, slab, *steps):
        return slab

    @staticmethod
    def backward(ctx, dtraj):
        return (None,) + tuple(dtraj[t] for t in range(dtraj.shape[0]))


class GroupedFFFn(torch.autograd.Function):
    """Grouped per-level MLP (d -> 4d -> GELU -> d), bottom-up or top-down.

    mode 0 (bottom-up): group 0 consumes `tokens`, group g consumes
    levels[..., g-1, :]  (reference glom_pytorch.py:132-134, without the cat)
    mode 1 (top-down): group g consumes levels[..., g+1, :] + pos; the pos
    add is materialized once per call by k_add_pos (glom_pytorch.py:136).
    """

    @staticmethod
    def forward(ctx, tokens, levels, pos, w1, b1, w2, b2, mode):
        ext = _load_extension()
        Y, Hpre, Hact, td_in = ext.grouped_ff_fwd(tokens, levels, pos, w1,
                                                  b1, w2, b2, mode)
        ctx.save_for_backward(tokens, levels, pos, w1, w2, Hpre, Hact,
                              td_in)
        ctx.mode = mode
        return Y

    @staticmethod
    def backward(ctx, dY):
        ext = _load_extension()
        tokens, levels, pos, w1, w2, Hpre, Hact, td_in = ctx.saved_tensors
        dTokens, dLevels, dW1, dB1, dW2, dB2 = ext.grouped_ff_bwd(
            dY.contiguous(), tokens, levels, pos, w1, w2, Hpre, Hact,
            ctx.mode, None, None, td_in)
        dPos = None
        if ctx.mode == 1 and pos is not None and ctx.needs_input_grad[2]:
            # pos was added to every top-down group input; its grad is the
            # sum of the level-slice grads over batch and groups (native
            # deterministic reduction).
            dPos = ext.dpos(dLevels)
        if ctx.mode == 0:
            return dTokens, dLevels, None, dW1, dB1, dW2, dB2, None
        return None, dLevels, dPos, dW1, dB1, dW2, dB2, None


class ConsensusFn(torch.autograd.Function):
    """Consensus attention across patch columns (glom_pytorch.py:38-73)."""

    @staticmethod
    def forward(ctx, levels, attend_self, mask):
        ext = _load_extension()
        out, probs, rnorm = ext.consensus_fwd(levels, attend_self, mask)
        ctx.save_for_backward(levels, probs, rnorm, mask)
        ctx.attend_self = attend_self
        return out

    @staticmethod
    def backward(ctx, dOut):
        ext = _load_extension()
        levels, probs, rnorm, mask = ctx.saved_tensors
        dLevels = ext.consensus_bwd(dOut.contiguous(), levels, probs, rnorm,
                                    ctx.attend_self, mask)
        return dLevels, None, None


class LevelMixFn(torch.autograd.Function):
    """(prev + bu + pad(td) + cons) / [4,..,4,3]  (glom_pytorch.py:141-144)."""

    @staticmethod
    def forward(ctx, levels, bu, td, cons):
        ext = _load_extension()
        return ext.level_mix_fwd(levels, bu, td, cons)

    @staticmethod
    def backward(ctx, dout):
        ext = _load_extension()
        dmix, dtd = ext.level_mix_bwd(dout.contiguous())
        # prev/bu/cons share the same scaled gradient; hand the engine
        # distinct view objects so it can never steal/accumulate into one
        # buffer in place on behalf of another input.
        return dmix, dmix.view_as(dmix), dtd, dmix.view_as(dmix)


class GlomStepFn(torch.autograd.Function):
    """One full GLOM iteration (reference glom_pytorch.py:131-145) as a
    single autograd node: forward chains bottom-up / top-down / consensus /
    level-mix in one extension call; backward sums the four levels-gradient
    contributions in one fused kernel instead of three engine
    accumulations."""

    _streams = None

    @staticmethod
    def _side_streams():
        if GlomStepFn._streams is None:
            GlomStepFn._streams = (torch.cuda.Stream(), torch.cuda.Stream(),
                                   torch.cuda.Stream())
        return GlomStepFn._streams

    @staticmethod
    def forward(ctx, tokens, levels, pos, bw1, bb1, bw2, bb2,
                tw1, tb1, tw2, tb2, attend_self, mask,
                bw1t, bw2t, tw1t, tw2t, slab_ref=None):
        # slab_ref: optional (slab, idx) — write the step output straight
        # into the preallocated trajectory slab (untracked alias return)
        ext = _load_extension()
        slab, sidx = slab_ref if slab_ref is not None else (None, 0)
        if (torch.is_grad_enabled()
                and os.environ.get("GLOM_FWD_STREAMS", "0") != "1"):
            (out, bhp, bha, thp, tha, probs, rnorm,
             tdin) = ext.glom_step_fwd(
                tokens, levels, pos, bw1, bb1, bw2, bb2, tw1, tb1, tw2,
                tb2, attend_self, mask, slab, sidx)
        else:
            # inference: the three per-iteration chains (bottom-up MLP,
            # top-down MLP, consensus attention) only depend on `levels`;
            # forking them onto side streams overlaps their tail waves
            # (+10% forward throughput). Training keeps one stream: the
            # backward-dominated step measured no gain from the fork.
            cur = torch.cuda.current_stream()
            s_td, s_at, _ = GlomStepFn._side_streams()
            ev = torch.cuda.Event()
            ev.record(cur)
            buY, bhp, bha, _ = ext.grouped_ff_fwd(tokens, levels, None,
                                                  bw1, bb1, bw2, bb2, 0)
            with torch.cuda.stream(s_td):
                s_td.wait_event(ev)
                tdY, thp, tha, tdin = ext.grouped_ff_fwd(None, levels, pos,
                                                         tw1, tb1, tw2,
                                                         tb2, 1)
            with torch.cuda.stream(s_at):
                s_at.wait_event(ev)
                cons, probs, rnorm = ext.consensus_fwd(levels, attend_self,
                                                       mask)
            cur.wait_stream(s_td)
            cur.wait_stream(s_at)
            # boundary tensors crossing back to the main stream: pin their
            # blocks until the main stream catches up (allocator safety)
            for t in (tdY, thp, tha, tdin, cons, probs, rnorm):
                t.record_stream(cur)
            out = ext.level_mix_fwd(levels, buY, tdY, cons, slab, sidx)
        ctx.save_for_backward(tokens, levels, pos, bw1, bw2, tw1, tw2,
                              bhp, bha, thp, tha, probs, rnorm, mask,
                              bw1t, bw2t, tw1t, tw2t, tdin)
        ctx.attend_self = attend_self
        return out

    @staticmethod
    def backward(ctx, dnew):
        ext = _load_extension()
        (tokens, levels, pos, bw1, bw2, tw1, tw2, bhp, bha, thp, tha,
         probs, rnorm, mask, bw1t, bw2t, tw1t, tw2t,
         tdin) = ctx.saved_tensors
        if os.environ.get("GLOM_NO_BWD_STREAMS", "0") == "1":
            (dTokens, dLevels, dPos, dbw1, dbb1, dbw2, dbb2,
             dtw1, dtb1, dtw2, dtb2) = ext.glom_step_bwd(
                dnew.contiguous(), tokens, levels, pos, bw1, bw2, tw1, tw2,
                bhp, bha, thp, tha, probs, rnorm, ctx.attend_self, mask,
                bw1t, bw2t, tw1t, tw2t, tdin)
            return (dTokens, dLevels, dPos, dbw1, dbb1, dbw2, dbb2,
                    dtw1, dtb1, dtw2, dtb2, None, None, None, None, None,
                    None, None)
        if os.environ.get("GLOM_BWD_FORK", "3") == "3":
            cur = torch.cuda.current_stream()
            s_td, s_at, _ = GlomStepFn._side_streams()
            dmix, dtd = ext.level_mix_bwd(dnew.contiguous())
            ev = torch.cuda.Event()
            ev.record(cur)
            bu = ext.grouped_ff_bwd(dmix, tokens, levels, None, bw1, bw2,
                                    bhp, bha, 0, bw1t, bw2t)
            with torch.cuda.stream(s_td):
                s_td.wait_event(ev)
                td = ext.grouped_ff_bwd(dtd, None, levels, pos, tw1, tw2,
                                        thp, tha, 1, tw1t, tw2t, tdin)
            with torch.cuda.stream(s_at):
                s_at.wait_event(ev)
                dAttn = ext.consensus_bwd(dmix, levels, probs, rnorm,
                                          ctx.attend_self, mask)
            dmix.record_stream(s_at)
            dtd.record_stream(s_td)
            cur.wait_stream(s_td)
            cur.wait_stream(s_at)
            for t in list(td) + [dAttn]:
                if t is not None and t.numel():
                    t.record_stream(cur)
            dLevels = torch.empty_like(levels)
            ext.add4_into(dmix, bu[1], td[1], dAttn, dLevels)
            dPos = ext.dpos(td[1])
            return (bu[0], dLevels, dPos, bu[2], bu[3], bu[4], bu[5],
                    td[2], td[3], td[4], td[5], None, None, None, None,
                    None, None, None)
        # 4-way fork variant (weight grads on a dedicated stream): measured
        # slightly SLOWER than the 3-way fork above (sync overhead), kept
        # behind GLOM_BWD_FORK=4 for future tuning.
        B, N, L = levels.size(0), levels.size(1), levels.size(2)
        cur = torch.cuda.current_stream()
        s_td, s_at, s_w = GlomStepFn._side_streams()
        dmix, dtd = ext.level_mix_bwd(dnew.contiguous())
        ev0 = torch.cuda.Event()
        ev0.record(cur)
        bu_dh, bu_db1 = ext.ff_bwd_dh(dmix, bw2t, bhp)
        ev_bu = torch.cuda.Event()
        ev_bu.record(cur)
        with torch.cuda.stream(s_td):
            s_td.wait_event(ev0)
            td_dh, td_db1 = ext.ff_bwd_dh(dtd, tw2t, thp)
            ev_td = torch.cuda.Event()
            ev_td.record(s_td)
        with torch.cuda.stream(s_at):
            s_at.wait_event(ev0)
            dAttn = ext.consensus_bwd(dmix, levels, probs, rnorm,
                                      ctx.attend_self, mask)
        with torch.cuda.stream(s_w):
            s_w.wait_event(ev_bu)
            bu_w1, bu_w2, bu_b2 = ext.ff_bwd_dw(dmix, bu_dh, tokens,
                                                levels, None, bha, 0)
            s_w.wait_event(ev_td)
            td_w1, td_w2, td_b2 = ext.ff_bwd_dw(dtd, td_dh, None, levels,
                                                pos, tha, 1, tdin)
        bu_dt, bu_dl = ext.ff_bwd_dx(bu_dh, bw1t, tokens, B, N, L, 0)
        with torch.cuda.stream(s_td):
            _, td_dl = ext.ff_bwd_dx(td_dh, tw1t, None, B, N, L, 1)
        # cross-stream block pinning
        for t, st in ((dmix, s_at), (dmix, s_w), (dtd, s_td), (dtd, s_w),
                      (bu_dh, s_w), (td_dh, s_w)):
            t.record_stream(st)
        cur.wait_stream(s_td)
        cur.wait_stream(s_at)
        cur.wait_stream(s_w)
        for t in (td_db1, td_dl, dAttn, bu_w1, bu_w2, bu_b2,
                  td_w1, td_w2, td_b2):
            t.record_stream(cur)
        dLevels = torch.empty_like(levels)
        ext.add4_into(dmix, bu_dl, td_dl, dAttn, dLevels)
        dPos = ext.dpos(td_dl)
        return (bu_dt, dLevels, dPos, bu_w1, bu_db1, bu_w2, bu_b2,
                td_w1, td_db1, td_w2, td_b2, None, None, None, None, None,
                None, None)


def _transposed_weights(model):
    """Per-group W^T copies for the backward NT GEMMs, computed ONCE per
    forward (they are loop constants across the T iterations)."""
    with torch.no_grad():
        L, d = model.levels, model.dim
        bw1 = model.bottom_up.net[1].weight[..., 0]
        bw2 = model.bottom_up.net[3].weight[..., 0]
        tw1 = model.top_down.net[1].weight[..., 0]
        tw2 = model.top_down.net[3].weight[..., 0]
        m4 = bw1.shape[0] // L
        return (
            bw1.view(L, m4, d).transpose(1, 2).contiguous(),
            bw2.view(L, d, m4).transpose(1, 2).contiguous(),
            tw1.view(L - 1, m4, d).transpose(1, 2).contiguous(),
            tw2.view(L - 1, d, m4).transpose(1, 2).contiguous(),
        )


def glom_step(model, tokens, levels, pos, mask, wts=None, slab_ref=None):
    bw = model.bottom_up.net
    tw = model.top_down.net
    if wts is None:
        wts = (None, None, None, None)
    return GlomStepFn.apply(
        tokens, levels, pos,
        bw[1].weight[..., 0], bw[1].bias, bw[3].weight[..., 0], bw[3].bias,
        tw[1].weight[..., 0], tw[1].bias, tw[3].weight[..., 0], tw[3].bias,
        model.attention.attend_self, mask, *wts, slab_ref)


_TAIL_STREAM = None


def _tail_stream():
    global _TAIL_STREAM
    if _TAIL_STREAM is None:
        _TAIL_STREAM = torch.cuda.Stream()
    return _TAIL_STREAM


def join_tail_stream():
    """Block the current stream on the overlap-tail stream. Callers that
    requested ``overlap_tail`` MUST call this before reading the
    trajectory's post-grad_iters slices or mutating model weights (the
    trainer joins right before the optimizer update)."""
    if _TAIL_STREAM is not None:
        torch.cuda.current_stream().wait_stream(_TAIL_STREAM)


def glom_forward(model, img, iters, levels=None, return_all=False,
                 grad_iters=None, overlap_tail=False):
    """grad_iters (optional): iterations >= grad_iters run under no_grad.
    Forward VALUES are identical; gradients stop flowing through those
    steps. When a loss only touches trajectory times <= grad_iters (the
    reference's denoising recipe decodes at t=7 of 12), the skipped
    backward contributions are exactly zero — autograd would otherwise
    backprop zeros through every post-loss iteration (reference
    glom_pytorch.py:131-148 has the same dead work)."""
    from glom_pytorch_amd.utils.profiling import trace_range
    b = img.shape[0]
    with trace_range("glom/patch_embed"):
        # K1: hand-written patchify + NT MFMA GEMM, once per forward
        emb = model.image_to_tokens[1]
        tokens = PatchEmbedFn.apply(img, emb.weight, emb.bias,
                                    model.patch_size)
    n = tokens.shape[1]
    pos = model.pos_emb.weight
    mask = (model.attention.non_local_mask
            if model.attention.local_consensus_radius > 0 else None)

    if levels is None:
        levels = model.init_levels.view(1, 1, model.levels, model.dim) \
            .expand(b, n, model.levels, model.dim).contiguous()
    else:
        levels = levels.contiguous()

    wts = _transposed_weights(model) if torch.is_grad_enabled() else None

    # return_all: steps write straight into a preallocated (T+1) slab —
    # no torch.stack copy at the end (SURVEY.md §2.3 note)
    slab = None
    if return_all:
        slab = torch.empty((iters + 1,) + tuple(levels.shape),
                           device=levels.device, dtype=levels.dtype)
        with torch.no_grad():
            slab[0].copy_(levels)

    # opt-in: run the forward-only tail (iterations >= grad_iters)
    # CONCURRENTLY with the caller's backward on a dedicated stream. The
    # tail never feeds the loss, and the caller joins (join_tail_stream)
    # before the optimizer mutates weights, so this is race-free.
    overlap = (overlap_tail and grad_iters is not None
               and grad_iters < iters and img.is_cuda
               and torch.is_grad_enabled())
    n_sync = grad_iters if overlap else iters

    steps = [levels]
    with trace_range(f"glom/iterate x{iters}"):
        for t in range(n_sync):
            sref = (slab, t + 1) if slab is not None else None
            with trace_range(f"glom/step{t}"):
                if grad_iters is not None and t >= grad_iters:
                    with torch.no_grad():
                        levels = glom_step(model, tokens, levels, pos,
                                           mask, wts, sref)
                else:
                    levels = glom_step(model, tokens, levels, pos, mask,
                                       wts, sref)
            if return_all:
                steps.append(levels)
        if overlap:
            ext = _load_extension()
            bw = model.bottom_up.net
            tw = model.top_down.net
            s_tail = _tail_stream()
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            with torch.no_grad(), torch.cuda.stream(s_tail):
                s_tail.wait_event(ev)
                for t in range(n_sync, iters):
                    out8 = ext.glom_step_fwd(
                        tokens, levels, pos,
                        bw[1].weight[..., 0], bw[1].bias,
                        bw[3].weight[..., 0], bw[3].bias,
                        tw[1].weight[..., 0], tw[1].bias,
                        tw[3].weight[..., 0], tw[3].bias,
                        model.attention.attend_self, mask, slab, t + 1)
                    levels = out8[0]
                    if return_all:
                        steps.append(levels)
                if slab is not None:
                    slab.record_stream(s_tail)

    if return_all:
        return TrajectoryFn.apply(slab, *steps)
    return levels
