"""Synthetic stand-ins for the MuJoCo continuous-control suite.

This image has no MuJoCo / dm_control, and the benchmark contract
(BASELINE.json) runs on synthetic env transitions anyway, so each MuJoCo
env name maps to a cheap deterministic surrogate with the REAL env's
observation/action dimensions.  Dynamics are a fixed random stable linear
map with tanh saturation plus process noise; reward mixes a state
component with an action-magnitude penalty so SAC has a learnable signal.

Shapes follow the MuJoCo v2/v4 envs the reference targets
(reference main.py:119 default Humanoid-v2; BASELINE configs 2/3/5 use
HalfCheetah-v4 / Humanoid-v4).
"""

import numpy as np

from .core import Box, Env


class SyntheticMujocoEnv(Env):
    """obs' = tanh(A obs + B a) + sigma*noise; r = w.obs - 0.05*|a|^2."""

    def __init__(self, obs_dim: int, act_dim: int, act_limit: float = 1.0,
                 ep_len: int = 1000, seed: int = 0, sigma: float = 0.01):
        self.obs_dim = obs_dim
        self.act_dim = act_dim
        self.ep_len = ep_len
        self.sigma = sigma
        self.action_space = Box(-act_limit, act_limit, (act_dim,))
        self.observation_space = Box(-np.inf, np.inf, (obs_dim,))
        init = np.random.default_rng(12345)  # fixed dynamics, not per-reset
        self._A = init.standard_normal((obs_dim, obs_dim)).astype(np.float32)
        self._A *= 0.9 / max(1.0, np.linalg.norm(self._A, 2))
        self._B = (init.standard_normal((obs_dim, act_dim)) / np.sqrt(act_dim)
                   ).astype(np.float32)
        self._w = init.standard_normal(obs_dim).astype(np.float32) / obs_dim
        self._rng = np.random.default_rng(seed)
        self._s = np.zeros(obs_dim, dtype=np.float32)
        self._t = 0

    def seed(self, seed: int):
        super().seed(seed)
        self._rng = np.random.default_rng(seed)

    def reset(self):
        self._s = 0.1 * self._rng.standard_normal(self.obs_dim).astype(np.float32)
        self._t = 0
        return self._s.copy()

    def step(self, action):
        a = np.clip(np.asarray(action, dtype=np.float32).reshape(self.act_dim),
                    self.action_space.low, self.action_space.high)
        s = np.tanh(self._A @ self._s + self._B @ a)
        if self.sigma > 0:
            s = s + self.sigma * self._rng.standard_normal(self.obs_dim).astype(np.float32)
        self._s = s.astype(np.float32)
        self._t += 1
        reward = float(self._w @ self._s) - 0.05 * float(a @ a)
        done = self._t >= self.ep_len
        return self._s.copy(), reward, done, {}


# (obs_dim, act_dim, act_limit) for the MuJoCo envs the reference suite uses
MUJOCO_SHAPES = {
    "HalfCheetah": (17, 6, 1.0),
    "Humanoid": (376, 17, 0.4),
    "Walker2d": (17, 6, 1.0),
    "Hopper": (11, 3, 1.0),
    "Ant": (27, 8, 1.0),
    "Swimmer": (8, 2, 1.0),
    "Reacher": (11, 2, 1.0),
    "InvertedPendulum": (4, 1, 3.0),
}


def make_factory(base: str):
    obs_dim, act_dim, act_limit = MUJOCO_SHAPES[base]
    def factory():
        return SyntheticMujocoEnv(obs_dim, act_dim, act_limit)
    return factory
