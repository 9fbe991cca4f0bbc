"""MLP actor / critic networks (MI355X-native).

API- and state-dict-compatible with the reference networks
(``networks/core.py:6-10``, ``networks/linear.py:13-79``): identical class
names, constructor signatures, parameter names (``layers.N.weight``,
``mu_layer``, ``log_std_layer``, ``q1``/``q2``) and output contracts.  The
forward paths run through ``torch_actor_critic_amd.ops`` — fused MFMA GEMM
+ tanh-Gaussian-head HIP kernels on gfx950, eager PyTorch on CPU.
"""

import typing as t

import torch
import torch.nn as nn

from ..ops import functional as Fo


def mlp(neurons: t.List[int]) -> nn.ModuleList:
    """A ModuleList of Linear layers from a width list; activations are
    applied by the caller (reference networks/core.py:6-10)."""
    return nn.ModuleList(
        [nn.Linear(n_in, n_out) for n_in, n_out in zip(neurons[:-1], neurons[1:])]
    )


class Actor(nn.Module):
    """Squashed-Gaussian policy (reference networks/linear.py:13-53).

    trunk: obs -> hidden (ReLU after every layer); heads: mu and log_std
    (clipped to [log_min_std, log_max_std]); action = tanh(sample)*act_limit
    with the numerically-stable tanh log-prob correction.
    """

    def __init__(
        self,
        obs_dim: int,
        act_dim: int,
        hidden_sizes: t.List[int],
        log_min_std: float = -20,
        log_max_std: float = 2,
        act_limit: float = 10,
    ):
        super().__init__()
        self.layers = mlp([obs_dim] + list(hidden_sizes))
        self.act_dim = act_dim
        self.mu_layer = nn.Linear(hidden_sizes[-1], act_dim)
        self.log_std_layer = nn.Linear(hidden_sizes[-1], act_dim)
        self.log_min_std = log_min_std
        self.log_max_std = log_max_std
        self.act_limit = act_limit

    def forward(self, x, deterministic: bool = False, with_logprob: bool = True):
        unbatched = x.ndim == 1
        if unbatched:
            x = x.unsqueeze(0)
        h = Fo.mlp_forward(x, self.layers, relu_last=True)
        mu = Fo.linear_relu(h, self.mu_layer.weight, self.mu_layer.bias, relu=False)
        log_std = Fo.linear_relu(h, self.log_std_layer.weight,
                                 self.log_std_layer.bias, relu=False)
        eps = Fo.randn_like_philox(mu) if not deterministic \
            else torch.zeros_like(mu)
        pi_action, logprob = Fo.tanh_gauss_head(
            mu, log_std, eps, self.act_limit, self.log_min_std,
            self.log_max_std, deterministic, with_logprob)
        if unbatched:
            pi_action = pi_action.squeeze(0)
            if logprob is not None:
                logprob = logprob.squeeze(0)
        return pi_action, logprob


class Critic(nn.Module):
    """Q(s,a): concat -> MLP -> scalar (reference networks/linear.py:56-69).
    ReLU on every layer except the final width-1 head; output squeezed."""

    def __init__(self, obs_dim: int, act_dim: int, hidden_sizes: t.List[int]):
        super().__init__()
        self.layers = mlp([obs_dim + act_dim] + list(hidden_sizes) + [1])

    def forward(self, state, action):
        x = torch.cat([state, action], dim=-1)
        unbatched = x.ndim == 1
        if unbatched:
            x = x.unsqueeze(0)
        x = Fo.mlp_forward(x, self.layers, relu_last=False)
        x = torch.squeeze(x, -1)
        if unbatched:
            x = x.squeeze(0)
        return x


class DoubleCritic(nn.Module):
    """Twin independent critics returning a tuple
    (reference networks/linear.py:72-79)."""

    def __init__(self, obs_dim: int, act_dim: int, hidden_sizes: t.List[int]):
        super().__init__()
        self.q1 = Critic(obs_dim, act_dim, hidden_sizes)
        self.q2 = Critic(obs_dim, act_dim, hidden_sizes)

    def forward(self, state, action):
        return self.q1(state, action), self.q2(state, action)
