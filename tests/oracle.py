"""Independent loop-level oracle for GLOM semantics.

Deliberately written as explicit per-level / per-step loops over basic torch
matmuls (no Conv1d, no einsum over 4-D, no einops) so it is an independent
re-derivation of the math in SURVEY.md §3.2 / §3.3, usable as ground truth
for both the eager path and the HIP engine.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

TOKEN_ATTEND_SELF_VALUE = -5e-4


def _group_weights(glom):
    """Slice the grouped-conv weights into per-level dense GEMM weights."""
    L, d, mult = glom.levels, glom.dim, glom.bottom_up.mult
    dm = d * mult

    def slices(ff, groups):
        w1 = ff.net[1].weight[..., 0]   # (G*dm, d)
        b1 = ff.net[1].bias
        w2 = ff.net[3].weight[..., 0]   # (G*d, dm)
        b2 = ff.net[3].bias
        out = []
        for g in range(groups):
            out.append((w1[g * dm:(g + 1) * dm], b1[g * dm:(g + 1) * dm],
                        w2[g * d:(g + 1) * d], b2[g * d:(g + 1) * d]))
        return out

    return slices(glom.bottom_up, L), slices(glom.top_down, L - 1)


def _mlp(x, w1, b1, w2, b2):
    h = x @ w1.t() + b1
    h = F.gelu(h)            # exact (erf) GELU
    return h @ w2.t() + b2


def oracle_forward(glom, img, iters=None, levels=None, return_all=False):
    if iters is None:
        iters = 2 * glom.levels
    L, d = glom.levels, glom.dim
    p = glom.patch_size
    b = img.shape[0]
    side = glom.num_patches_side
    n = glom.num_patches

    # patchify by explicit loops: token t = (h, w) patch, flattened (p1 p2 c)
    toks = []
    for hh in range(side):
        for ww in range(side):
            patch = img[:, :, hh * p:(hh + 1) * p, ww * p:(ww + 1) * p]
            toks.append(patch.permute(0, 2, 3, 1).reshape(b, -1))
    tokens = torch.stack(toks, dim=1)  # (b, n, p*p*3)
    lin = glom.image_to_tokens[1]
    tokens = tokens @ lin.weight.t() + lin.bias  # (b, n, d)

    pos = glom.pos_emb.weight[:n]  # (n, d)

    if levels is None:
        levels = glom.init_levels.view(1, 1, L, d).expand(b, n, L, d).clone()

    bu_w, td_w = _group_weights(glom)

    traj = [levels]
    for _ in range(iters):
        new = torch.empty_like(levels)
        # consensus attention, per batch & level
        cons = torch.empty_like(levels)
        for bi in range(b):
            for li in range(L):
                x = levels[bi, :, li, :]                       # (n, d)
                k = F.normalize(x, dim=-1)
                s = (x @ k.t()) / math.sqrt(d)
                if not glom.attention.attend_self:
                    s = s.clone()
                    s.fill_diagonal_(TOKEN_ATTEND_SELF_VALUE)
                if glom.attention.local_consensus_radius > 0:
                    m = glom.attention.non_local_mask[0]
                    s = s.masked_fill(m, -torch.finfo(s.dtype).max)
                a = s.softmax(dim=-1)
                cons[bi, :, li, :] = a @ x

        for li in range(L):
            below = tokens if li == 0 else levels[:, :, li - 1, :]
            bu = _mlp(below.reshape(b * n, d), *bu_w[li]).view(b, n, d)
            if li < L - 1:
                above = levels[:, :, li + 1, :] + pos.view(1, n, d)
                td = _mlp(above.reshape(b * n, d), *td_w[li]).view(b, n, d)
                c = 4.0
            else:
                td = torch.zeros(b, n, d, dtype=levels.dtype)
                c = 3.0
            new[:, :, li, :] = (levels[:, :, li, :] + bu + td
                                + cons[:, :, li, :]) / c
        levels = new
        traj.append(levels)

    if return_all:
        return torch.stack(traj)
    return levels
