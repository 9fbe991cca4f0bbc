"""Unit tests for utils: normalizer math, throughput counters, roctx
no-op safety, collective timing counters."""

import time

import numpy as np
import torch

from sac.utils import Identity, WelfordVarianceEstimate
from torch_actor_critic_amd.parallel import comm
from torch_actor_critic_amd.utils.profiling import Throughput, roctx_range


def test_welford_matches_numpy():
    rng = np.random.default_rng(0)
    data = rng.standard_normal((200, 5)) * 3 + 1
    w = WelfordVarianceEstimate()
    for row in data:
        w.update(torch.as_tensor(row))
    mean = w.mean.numpy()
    var = (w.m2 / (w.count - 1)).numpy()
    np.testing.assert_allclose(mean, data.mean(0), rtol=1e-6)
    np.testing.assert_allclose(var, data.var(0, ddof=1), rtol=1e-6)
    x = torch.as_tensor(data[0])
    nx = w.normalize_state(x).numpy()
    np.testing.assert_allclose(nx, (data[0] - mean) / np.sqrt(var + 1e-8),
                               rtol=1e-5)
    sd = w.state_dict()
    w2 = WelfordVarianceEstimate()
    w2.load_state_dict(sd)
    np.testing.assert_allclose(w2.normalize_state(x).numpy(), nx)


def test_identity_normalizer():
    x = torch.randn(4)
    assert torch.equal(Identity().normalize_state(x), x)


def test_throughput_counters():
    thr = Throughput()
    thr.tick_env(10)
    thr.tick_update(5)
    time.sleep(0.01)
    r = thr.rates()
    assert r["env_steps_per_sec"] > 0
    assert r["updates_per_sec"] > 0
    assert r["env_steps_per_sec"] / r["updates_per_sec"] == 2.0
    thr.reset()
    assert thr.env_steps == 0


def test_roctx_range_noop():
    with roctx_range("test_phase"):
        pass  # must not crash with or without roctx present


def test_collective_stats_counters():
    # not initialized: counters stay zero, API still works
    cs = comm.collective_stats()
    assert cs["allreduce_n"] == 0
    comm.allreduce_grads(torch.zeros(4))  # no-op without dist
    cs = comm.collective_stats()
    assert cs["allreduce_n"] == 0
