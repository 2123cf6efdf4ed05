"""Fused AdamW for bf16 training with fp32 master weights.

One HIP kernel pass per step (plus a two-stage deterministic grad-norm
reduction) replaces the per-step Python loop of: bf16->fp32 grad copies,
clip_grad_norm_, foreach-AdamW over the masters, and the fp32->bf16
parameter copy-back. Semantics match ``torch.optim.AdamW`` exactly
(decoupled weight decay, bias correction, the clip applied to grads before
the moment updates), and ``state_dict()`` mirrors torch's format so
checkpoints interchange with the plain-optimizer path.

hipGraph-replay-safe: the step counter lives in a device tensor that the
norm-finalize kernel increments, so a captured step() stays correct across
replays; ``steps_done()`` syncs the host mirror (one device read) before
checkpointing.

Per-param step caveat: torch increments a param's ``step`` only on steps
where it received a gradient; this class uses the global step count for
every present param's bias correction (params with ``grad=None`` are
skipped entirely, like torch). Identical whenever all params get grads
every step — the standard training path.

Hyperparameter caveat under graph capture: ``lr``/betas/eps/weight_decay
are kernel arguments, so a captured training step bakes them in. To
change them mid-run, update the attribute AND drop the captured graphs
(``trainer._graphs.clear()``) so the next step re-captures.
"""

from __future__ import annotations

import torch

from glom_pytorch_amd.ops import _load_extension

_NPART = 1024   # must match OPT_NPART in native_ops.h


class FusedAdamW:
    def __init__(self, params: list[torch.Tensor],
                 masters: list[torch.Tensor], *, lr: float = 1e-3,
                 betas: tuple[float, float] = (0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 1e-2,
                 max_grad_norm: float = 0.0):
        assert len(params) == len(masters)
        self.params = list(params)
        self.masters = list(masters)
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.max_grad_norm = max_grad_norm
        dev = masters[0].device
        self.exp_avg = [torch.zeros_like(m) for m in masters]
        self.exp_avg_sq = [torch.zeros_like(m) for m in masters]
        self._partials = torch.zeros(_NPART, dtype=torch.float32, device=dev)
        self._norm = torch.zeros(1, dtype=torch.float32, device=dev)
        self._step_dev = torch.zeros(1, dtype=torch.float32, device=dev)
        self._step_host = 0      # mirror; device tensor is authoritative

    def zero_grad(self, set_to_none: bool = True):
        for p in self.params:
            p.grad = None

    @torch.no_grad()
    def step(self):
        ext = _load_extension()
        gs, mws, m1s, m2s, pws = [], [], [], [], []
        for i, p in enumerate(self.params):
            if p.grad is None:
                continue
            gs.append(p.grad)
            mws.append(self.masters[i])
            m1s.append(self.exp_avg[i])
            m2s.append(self.exp_avg_sq[i])
            pws.append(p.data)
        if not gs:
            return
        ext.fused_adamw(gs, mws, m1s, m2s, pws, self.lr, self.betas[0],
                        self.betas[1], self.eps, self.weight_decay,
                        self.max_grad_norm, self._partials, self._norm,
                        self._step_dev)
        self._step_host += 1

    def note_replays(self, n: int):
        """Called by graph-replay drivers: n captured steps ran without
        Python; keep the host mirror roughly in sync (device is exact)."""
        self._step_host += n

    def steps_done(self) -> int:
        """Exact step count (reads the device counter; syncs)."""
        return int(self._step_dev.item())

    # ----------------- torch.optim.AdamW state_dict parity -------------

    def state_dict(self):
        step = float(self.steps_done())
        state = {}
        for i in range(len(self.params)):
            state[i] = {
                "step": torch.tensor(step),
                "exp_avg": self.exp_avg[i],
                "exp_avg_sq": self.exp_avg_sq[i],
            }
        group = {
            "lr": self.lr, "betas": tuple(self.betas), "eps": self.eps,
            "weight_decay": self.weight_decay, "amsgrad": False,
            "foreach": None, "maximize": False, "capturable": False,
            "differentiable": False, "fused": None,
            "params": list(range(len(self.params))),
        }
        return {"state": state, "param_groups": [group]}

    def load_state_dict(self, sd):
        g = sd["param_groups"][0]
        self.lr = g["lr"]
        self.betas = tuple(g["betas"])
        self.eps = g["eps"]
        self.weight_decay = g["weight_decay"]
        step = 0.0
        for i, m in enumerate(self.masters):
            st = sd["state"].get(i, sd["state"].get(str(i)))
            if st is None:
                continue
            self.exp_avg[i].copy_(st["exp_avg"].to(m.device,
                                                   torch.float32))
            self.exp_avg_sq[i].copy_(st["exp_avg_sq"].to(m.device,
                                                         torch.float32))
            s = st["step"]
            step = float(s.item() if torch.is_tensor(s) else s)
        self._step_dev.fill_(step)
        self._step_host = int(step)
