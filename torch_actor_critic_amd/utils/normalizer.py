"""State normalizers (parity with the reference's sac/utils.py:10-79).

The reference ships these but never wires them in (dead code — SURVEY.md
Q9).  They are provided here for capability parity and CAN be wired in
via ``main.py --normalize-states`` (deliberate, documented extension;
default off for reference-faithful behavior).
"""

import abc

import torch

from . import checkpoint as ckpt


class StateNormalizer(abc.ABC):
    @abc.abstractmethod
    def normalize_state(self, state: torch.Tensor) -> torch.Tensor:
        ...

    def save(self, artifact_path: str = "normalizer"):
        ckpt.log_state_dict(self.state_dict(), artifact_path)

    # reference API names (sac/utils.py:19-24): save_state logs the
    # state dict to an artifact path, load_state restores from one
    def save_state(self, path: str = "normalizer"):
        self.save(path)

    def load_state(self, state_dict: dict):
        self.load_state_dict(state_dict)

    def state_dict(self) -> dict:
        return {}

    def load_state_dict(self, sd: dict):
        pass


class Identity(StateNormalizer):
    def normalize_state(self, state: torch.Tensor) -> torch.Tensor:
        return state


class WelfordVarianceEstimate(StateNormalizer):
    """Online mean/variance (Welford) normalization."""

    def __init__(self, eps: float = 1e-8):
        self.count = 0
        self.mean = None
        self.m2 = None
        self.eps = eps

    def update(self, state: torch.Tensor):
        x = state.detach().to(torch.float64)
        if self.mean is None:
            self.mean = torch.zeros_like(x)
            self.m2 = torch.zeros_like(x)
        self.count += 1
        delta = x - self.mean
        self.mean += delta / self.count
        self.m2 += delta * (x - self.mean)

    def normalize_state(self, state: torch.Tensor) -> torch.Tensor:
        if self.mean is None or self.count < 2:
            return state
        var = self.m2 / (self.count - 1)
        return ((state.to(torch.float64) - self.mean)
                / torch.sqrt(var + self.eps)).to(state.dtype)

    def state_dict(self) -> dict:
        return {"count": self.count, "mean": self.mean, "m2": self.m2}

    def load_state_dict(self, sd: dict):
        self.count = sd["count"]
        self.mean = sd["mean"]
        self.m2 = sd["m2"]
