"""Compat shape tests for the visual networks (mirrors the reference's
tests/test_convolutional.py, including the unbatched auto-unsqueeze /
squeeze contract — reference convolutional.py:91-96,121,147-154)."""

import torch

from networks.convolutional import (MultiObservation, VisualActor,
                                    VisualCritic, VisualDoubleCritic,
                                    calculate_size, simple_cnn)


def _obs(batch=None, feat=24, vis=(3, 64, 64)):
    if batch is None:
        return MultiObservation(torch.randn(feat), torch.randn(*vis))
    return MultiObservation(torch.randn(batch, feat),
                            torch.randn(batch, *vis))


def test_calculate_size_nature_cnn():
    # 3x64x64 -> conv 32@8s4 -> 64@4s2 -> 64@3s1 => 64*4*4 = 1024
    assert calculate_size((3, 64, 64), [32, 64, 64], [8, 4, 3],
                          [4, 2, 1]) == 1024


def test_simple_cnn_output_scalar():
    net = simple_cnn((3, 64, 64))
    out = net(torch.randn(2, 3, 64, 64))
    assert out.shape == (2, 1)


def test_visual_actor_unbatched():
    actor = VisualActor(obs_dim=24, act_dim=5, vis_dim=(3, 64, 64),
                        hidden_sizes=[32, 32])
    pi, logp = actor(_obs())
    assert pi.shape == (5,)
    assert logp.shape == ()


def test_visual_actor_batched():
    actor = VisualActor(obs_dim=24, act_dim=5, vis_dim=(3, 64, 64),
                        hidden_sizes=[32, 32])
    pi, logp = actor(_obs(batch=4))
    assert pi.shape == (4, 5)
    assert logp.shape == (4,)


def test_visual_critic_batched_and_unbatched():
    critic = VisualCritic(obs_dim=24, act_dim=5, vis_dim=(3, 64, 64),
                          hidden_sizes=[32, 32])
    q = critic(_obs(batch=4), torch.randn(4, 5))
    assert q.shape == (4,)
    q = critic(_obs(), torch.randn(5))
    assert q.shape == (1,)  # reference auto-unsqueeze keeps batch dim of 1


def test_visual_double_critic():
    critic = VisualDoubleCritic(obs_dim=24, act_dim=5, vis_dim=(3, 64, 64),
                                hidden_sizes=[32])
    q1, q2 = critic(_obs(batch=3), torch.randn(3, 5))
    assert q1.shape == (3,) and q2.shape == (3,)
    assert not torch.allclose(q1, q2)


def test_visual_gradients_flow():
    actor = VisualActor(obs_dim=12, act_dim=3, vis_dim=(3, 64, 64),
                        hidden_sizes=[16])
    pi, logp = actor(_obs(batch=2, feat=12))
    (pi.sum() + logp.sum()).backward()
    for p in actor.parameters():
        assert p.grad is not None
