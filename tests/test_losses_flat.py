"""Numerics tests: SAC losses vs independent formulas; FlatParams /
FlatAdam vs torch.optim.Adam; polyak vs per-parameter loop."""

import numpy as np
import torch
import torch.nn as nn

from networks.linear import Actor, DoubleCritic
from sac.algorithm import eval_pi_loss, eval_q_loss, update_targets
from torch_actor_critic_amd.ops import functional as Fo
from torch_actor_critic_amd.optim import FlatAdam
from torch_actor_critic_amd.parallel.flat import FlatParams, flatten_module_like


def test_q_loss_matches_manual():
    B = 32
    q1, q2 = torch.randn(B), torch.randn(B)
    q1t, q2t = torch.randn(B), torch.randn(B)
    logp = torch.randn(B)
    r = torch.randn(B)
    d = (torch.rand(B) > 0.7).float()
    alpha, gamma, scale = 0.2, 0.99, 1.5

    loss = Fo.sac_q_loss(q1, q2, q1t, q2t, logp, r, d, alpha, gamma, scale)
    backup = scale * r + gamma * (1 - d) * (torch.min(q1t, q2t) - alpha * logp)
    expected = ((q1 - backup) ** 2).mean() + ((q2 - backup) ** 2).mean()
    assert torch.allclose(loss, expected, atol=1e-6)


def test_q_loss_gradients():
    B = 16
    q1 = torch.randn(B, requires_grad=True)
    q2 = torch.randn(B, requires_grad=True)
    loss = Fo.sac_q_loss(q1, q2, torch.randn(B), torch.randn(B),
                         torch.randn(B), torch.randn(B),
                         torch.zeros(B), 0.2, 0.99, 1.0)
    loss.backward()
    # d/dq1 mse(q1, backup) = 2*(q1-backup)/B
    assert q1.grad is not None and torch.isfinite(q1.grad).all()


def test_pi_loss_matches_manual():
    B = 32
    q1, q2, logp = torch.randn(B), torch.randn(B), torch.randn(B)
    loss = Fo.sac_pi_loss(q1, q2, logp, 0.2)
    expected = (0.2 * logp - torch.min(q1, q2)).mean()
    assert torch.allclose(loss, expected, atol=1e-6)


def test_eval_losses_end_to_end():
    actor = Actor(5, 2, [16, 16], act_limit=1.0)
    critic = DoubleCritic(5, 2, [16, 16])
    target = DoubleCritic(5, 2, [16, 16])
    s, ns = torch.randn(8, 5), torch.randn(8, 5)
    a = torch.randn(8, 2)
    r, d = torch.randn(8), torch.zeros(8)

    lq = eval_q_loss(actor, critic, target, s, a, r, ns, d, 0.2, 0.99, 1.0)
    assert lq.ndim == 0 and lq.item() >= 0
    lp = eval_pi_loss(actor, critic, s, ns, 0.2)
    assert torch.isfinite(lp)


def test_update_targets_polyak():
    src = nn.Linear(4, 4)
    targ = nn.Linear(4, 4)
    t0 = targ.weight.data.clone()
    update_targets(src, targ, 0.9)
    expected = 0.9 * t0 + 0.1 * src.weight.data
    assert torch.allclose(targ.weight.data, expected, atol=1e-6)


def test_flat_polyak_matches_looped():
    src = DoubleCritic(4, 2, [8, 8])
    targ_a = DoubleCritic(4, 2, [8, 8])
    targ_b = DoubleCritic(4, 2, [8, 8])
    targ_b.load_state_dict(targ_a.state_dict())

    update_targets(src, targ_a, 0.995)

    src_flat = FlatParams(src)
    targ_flat = flatten_module_like(targ_b)
    Fo.polyak_(targ_flat, src_flat.flat, 0.995)
    for pa, pb in zip(targ_a.parameters(), targ_b.parameters()):
        assert torch.allclose(pa.data, pb.data, atol=1e-6)


def test_flat_params_views_survive_backward():
    """Autograd must accumulate into the flat grad bucket — the single-
    bucket all-reduce depends on it."""
    actor = Actor(5, 2, [16], act_limit=1.0)
    fp = FlatParams(actor)
    fp.zero_grad()
    pi, logp = actor(torch.randn(8, 5))
    (pi.sum() + logp.sum()).backward()
    assert fp.check_views(), "autograd replaced a .grad view"
    assert fp.flat_grad.abs().sum() > 0


def test_flat_adam_matches_torch_adam():
    torch.manual_seed(0)
    m1 = nn.Sequential(nn.Linear(6, 8), nn.ReLU(), nn.Linear(8, 1))
    m2 = nn.Sequential(nn.Linear(6, 8), nn.ReLU(), nn.Linear(8, 1))
    m2.load_state_dict(m1.state_dict())

    opt1 = torch.optim.Adam(m1.parameters(), lr=1e-3)
    opt2 = FlatAdam(m2, lr=1e-3)

    x = torch.randn(32, 6)
    y = torch.randn(32, 1)
    for _ in range(5):
        opt1.zero_grad()
        nn.functional.mse_loss(m1(x), y).backward()
        opt1.step()

        opt2.zero_grad()
        nn.functional.mse_loss(m2(x), y).backward()
        opt2.step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.data, p2.data, atol=1e-6), \
            (p1.data - p2.data).abs().max()


def test_flat_adam_state_dict_roundtrip():
    m = nn.Linear(4, 4)
    opt = FlatAdam(m, lr=1e-3)
    m(torch.randn(2, 4)).sum().backward()
    opt.step()
    sd = opt.state_dict()
    assert sd["param_groups"][0]["lr"] == 1e-3
    assert 0 in sd["state"] and "exp_avg" in sd["state"][0]

    m2 = nn.Linear(4, 4)
    m2.load_state_dict(m.state_dict())
    opt2 = FlatAdam(m2, lr=5e-4)
    opt2.load_state_dict(sd)
    assert opt2.lr == 1e-3
    assert int(opt2.step_t.item()) == 1
    assert torch.allclose(opt2.m, opt.m)


def test_flat_adam_interchanges_with_torch_adam_state():
    """FlatAdam state dict loads into torch.optim.Adam and vice versa
    (checkpoint interchange with the reference's auxiliaries format)."""
    m = nn.Linear(4, 2)
    fa = FlatAdam(m, lr=1e-3)
    m(torch.randn(3, 4)).sum().backward()
    fa.step()
    sd = fa.state_dict()

    m2 = nn.Linear(4, 2)
    ta = torch.optim.Adam(m2.parameters(), lr=1e-3)
    ta.load_state_dict(sd)  # must not raise

    fa2 = FlatAdam(nn.Linear(4, 2), lr=1e-3)
    fa2.load_state_dict(ta.state_dict())  # and back
