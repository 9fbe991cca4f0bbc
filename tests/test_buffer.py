"""Replay buffer property tests (store/sample/ring semantics —
reference buffer/replay_buffer.py had no tests at all)."""

import numpy as np
import torch

from buffer.replay_buffer import Batch, ReplayBuffer
from buffer.visual_replay_buffer import VisualBatch, VisualReplayBuffer
from torch_actor_critic_amd.envs.visual import MultiObservation


def _fill(buf, n, obs_dim=4, act_dim=2):
    for i in range(n):
        buf.store(np.full(obs_dim, i, dtype=np.float32),
                  np.full(act_dim, i, dtype=np.float32),
                  float(i), np.full(obs_dim, i + 1, dtype=np.float32),
                  float(i % 2))


def test_store_and_sample_shapes():
    buf = ReplayBuffer(100, 4, 2)
    _fill(buf, 10)
    batch = buf.sample(8)
    assert isinstance(batch, Batch)
    assert batch.states.shape == (8, 4)
    assert batch.actions.shape == (8, 2)
    assert batch.rewards.shape == (8,)
    assert batch.next_states.shape == (8, 4)
    assert batch.done.shape == (8,)
    assert batch.states.dtype == torch.float32


def test_sample_row_consistency():
    """Each sampled row must keep its transition fields together."""
    buf = ReplayBuffer(50, 4, 2)
    _fill(buf, 30)
    batch = buf.sample(16)
    for j in range(16):
        i = batch.states[j, 0].item()
        assert (batch.states[j] == i).all()
        assert (batch.actions[j] == i).all()
        assert batch.rewards[j].item() == i
        assert (batch.next_states[j] == i + 1).all()
        assert batch.done[j].item() == float(int(i) % 2)


def test_ring_wraparound():
    buf = ReplayBuffer(8, 4, 2)
    _fill(buf, 20)
    assert buf.size == 8
    assert buf.ptr == 20 % 8
    # oldest entries overwritten: only values 12..19 remain
    vals = buf.state[:, 0]
    assert vals.min().item() >= 12


def test_store_batch_matches_store():
    b1 = ReplayBuffer(100, 3, 2)
    b2 = ReplayBuffer(100, 3, 2)
    rng = np.random.default_rng(0)
    obs = rng.standard_normal((17, 3)).astype(np.float32)
    act = rng.standard_normal((17, 2)).astype(np.float32)
    rew = rng.standard_normal(17).astype(np.float32)
    nxt = rng.standard_normal((17, 3)).astype(np.float32)
    done = (rng.random(17) > 0.5).astype(np.float32)
    for i in range(17):
        b1.store(obs[i], act[i], rew[i], nxt[i], done[i])
    b2.store_batch(obs, act, rew, nxt, done)
    assert b1.size == b2.size and b1.ptr == b2.ptr
    assert torch.allclose(b1.state, b2.state)
    assert torch.allclose(b1.rewards, b2.rewards)
    assert torch.allclose(b1.done, b2.done)


def test_store_batch_wraps():
    buf = ReplayBuffer(10, 3, 2)
    obs = np.arange(24 * 3, dtype=np.float32).reshape(24, 3)
    buf.store_batch(obs, np.zeros((24, 2)), np.zeros(24), obs, np.zeros(24))
    assert buf.size == 10
    assert buf.ptr == 24 % 10


def test_visual_buffer_roundtrip():
    buf = VisualReplayBuffer(20, act_dim=3, quantize_frames=False)
    for i in range(6):
        mo = MultiObservation(torch.full((7,), float(i)),
                              torch.full((3, 8, 8), float(i) / 10))
        buf.store(mo, np.zeros(3), float(i), mo, 0.0)
    batch = buf.sample(4)
    assert isinstance(batch, VisualBatch)
    assert batch.states.features.shape == (4, 7)
    assert batch.states.frame.shape == (4, 3, 8, 8)
    j = batch.states.features[0, 0].item()
    assert torch.allclose(batch.states.frame[0],
                          torch.full((3, 8, 8), j / 10))


def test_visual_buffer_quantization_error_bounded():
    buf = VisualReplayBuffer(10, act_dim=2, quantize_frames=True)
    frame = torch.rand(3, 8, 8) * 2 - 1
    mo = MultiObservation(torch.zeros(4), frame)
    buf.store(mo, np.zeros(2), 0.0, mo, 0.0)
    batch = buf.sample(1)
    err = (batch.states.frame[0] - frame).abs().max().item()
    assert err <= 1.0 / 127.5  # one quantization step
