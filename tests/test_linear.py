"""Shape-contract compat suite for the MLP networks (mirrors the
reference's tests/test_linear.py) plus numerics tests of the fused ops'
eager contract against independent formulas."""

import math

import numpy as np
import pytest
import torch
import torch.nn.functional as F
from torch.distributions.normal import Normal

from networks.linear import Actor, Critic, DoubleCritic


def test_actor_unbatched_shapes():
    actor = Actor(obs_dim=8, act_dim=3, hidden_sizes=[32, 32], act_limit=2.0)
    obs = torch.randn(8)
    pi, logp = actor(obs)
    assert pi.shape == (3,)
    assert logp.shape == ()
    assert torch.all(pi.abs() <= 2.0 + 1e-5)


def test_actor_batched_shapes():
    actor = Actor(obs_dim=8, act_dim=3, hidden_sizes=[32, 32])
    obs = torch.randn(16, 8)
    pi, logp = actor(obs)
    assert pi.shape == (16, 3)
    assert logp.shape == (16,)


def test_actor_deterministic_no_logprob():
    actor = Actor(obs_dim=4, act_dim=2, hidden_sizes=[16])
    obs = torch.randn(5, 4)
    pi, logp = actor(obs, deterministic=True, with_logprob=False)
    assert pi.shape == (5, 2)
    assert logp is None
    pi2, _ = actor(obs, deterministic=True, with_logprob=False)
    assert torch.allclose(pi, pi2)


def test_actor_logprob_matches_distributions_formula():
    """The fused head contract must equal the reference composite built
    from torch.distributions (reference networks/linear.py:37-53)."""
    torch.manual_seed(3)
    actor = Actor(obs_dim=6, act_dim=4, hidden_sizes=[32], act_limit=1.5)
    obs = torch.randn(32, 6)
    torch.manual_seed(7)
    pi, logp = actor(obs)

    # independent recomputation
    x = obs
    for layer in actor.layers:
        x = F.relu(layer(x))
    mu = actor.mu_layer(x)
    log_std = torch.clip(actor.log_std_layer(x), -20, 2)
    std = torch.exp(log_std)
    torch.manual_seed(7)
    eps = torch.randn_like(mu)
    prob = mu + std * eps
    dist = Normal(mu, std)
    ref_logp = dist.log_prob(prob).sum(-1)
    ref_logp = ref_logp - (2 * math.log(2) - prob
                           - F.softplus(-2 * prob)).sum(-1)
    ref_pi = torch.tanh(prob) * 1.5

    assert torch.allclose(pi, ref_pi, atol=1e-5)
    assert torch.allclose(logp, ref_logp, atol=1e-4)


def test_critic_shapes():
    critic = Critic(obs_dim=8, act_dim=3, hidden_sizes=[32, 32])
    q = critic(torch.randn(16, 8), torch.randn(16, 3))
    assert q.shape == (16,)


def test_critic_unbatched():
    critic = Critic(obs_dim=8, act_dim=3, hidden_sizes=[32])
    q = critic(torch.randn(8), torch.randn(3))
    assert q.shape == ()


def test_double_critic_shapes():
    critic = DoubleCritic(obs_dim=8, act_dim=3, hidden_sizes=[32, 32])
    q1, q2 = critic(torch.randn(16, 8), torch.randn(16, 3))
    assert q1.shape == (16,)
    assert q2.shape == (16,)
    # twin critics must be independent
    assert not torch.allclose(q1, q2)


def test_state_dict_layout_matches_reference_names():
    """Parameter names must interchange with reference checkpoints
    (layers.N.weight/bias, mu_layer, log_std_layer, q1/q2)."""
    actor = Actor(obs_dim=4, act_dim=2, hidden_sizes=[8, 8])
    keys = set(actor.state_dict().keys())
    assert {"layers.0.weight", "layers.0.bias", "layers.1.weight",
            "layers.1.bias", "mu_layer.weight", "mu_layer.bias",
            "log_std_layer.weight", "log_std_layer.bias"} == keys

    critic = DoubleCritic(obs_dim=4, act_dim=2, hidden_sizes=[8])
    keys = set(critic.state_dict().keys())
    assert "q1.layers.0.weight" in keys and "q2.layers.1.bias" in keys


def test_gradients_flow():
    actor = Actor(obs_dim=5, act_dim=2, hidden_sizes=[16, 16])
    pi, logp = actor(torch.randn(8, 5))
    (pi.sum() + logp.sum()).backward()
    for p in actor.parameters():
        assert p.grad is not None
        assert torch.isfinite(p.grad).all()
