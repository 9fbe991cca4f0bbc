"""FlatAdam — fused Adam over a module's flat parameter buffer.

One kernel launch per step on GPU (vs torch.optim.Adam's per-tensor
loop), hipGraph-replayable (the step counter lives in device memory and
is incremented by the kernel).  State-dict compatible with
``torch.optim.Adam`` so checkpoints interchange with the reference's
auxiliaries format (reference main.py:38-42, sac/algorithm.py:176-180).
"""

import typing as t

import torch

from .ops import functional as Fo
from .parallel.flat import FlatParams


class FlatAdam:
    def __init__(self, flat: t.Union[FlatParams, torch.nn.Module],
                 lr: float = 3e-4, betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        if isinstance(flat, torch.nn.Module):
            flat = FlatParams(flat)
        self.fp = flat
        self.lr = lr
        self.betas = tuple(betas)
        self.eps = eps
        self.weight_decay = weight_decay
        dev = flat.flat.device
        self.m = torch.zeros_like(flat.flat)
        self.v = torch.zeros_like(flat.flat)
        self.step_t = torch.zeros(1, dtype=torch.int64, device=dev)

    # -- torch.optim-style API -----------------------------------------

    def zero_grad(self, set_to_none: bool = False):  # noqa: ARG002
        self.fp.zero_grad()

    @torch.no_grad()
    def step(self):
        Fo.adam_step_(self.fp.flat, self.fp.flat_grad, self.m, self.v,
                      self.step_t, self.lr, self.betas[0], self.betas[1],
                      self.eps, self.weight_decay)

    # -- checkpointing (torch.optim.Adam-compatible layout) ------------

    def state_dict(self) -> dict:
        step = int(self.step_t.item())
        state = {}
        for i, (off, n) in enumerate(self.fp._slices):
            p = self.fp._params[i]
            state[i] = {
                "step": torch.tensor(float(step)),
                "exp_avg": self.m[off:off + n].view_as(p).clone().cpu(),
                "exp_avg_sq": self.v[off:off + n].view_as(p).clone().cpu(),
            }
        return {
            "state": state,
            "param_groups": [{
                "lr": self.lr, "betas": list(self.betas), "eps": self.eps,
                "weight_decay": self.weight_decay, "amsgrad": False,
                "maximize": False, "foreach": None, "capturable": False,
                "differentiable": False, "fused": None,
                "params": list(range(len(self.fp._params))),
            }],
        }

    def load_state_dict(self, sd: dict):
        groups = sd.get("param_groups")
        if groups:
            g = groups[0]
            self.lr = g.get("lr", self.lr)
            self.betas = tuple(g.get("betas", self.betas))
            self.eps = g.get("eps", self.eps)
            self.weight_decay = g.get("weight_decay", self.weight_decay)
        state = sd.get("state", {})
        step = 0
        for i, (off, n) in enumerate(self.fp._slices):
            s = state.get(i) or state.get(str(i))
            if s is None:
                continue
            self.m[off:off + n].copy_(
                torch.as_tensor(s["exp_avg"]).reshape(-1).to(self.m.device))
            self.v[off:off + n].copy_(
                torch.as_tensor(s["exp_avg_sq"]).reshape(-1).to(self.v.device))
            st = s.get("step", 0)
            step = int(st.item() if torch.is_tensor(st) else st)
        self.step_t.fill_(step)

    @property
    def param_groups(self):
        return [{"lr": self.lr, "params": self.fp._params}]
