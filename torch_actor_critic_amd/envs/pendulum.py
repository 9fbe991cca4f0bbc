"""Pendulum-v1 with the real classic-control dynamics, implemented
natively (no gym dependency).  This is the learning-curve sanity
environment: a correct SAC implementation takes
episode reward from ~-1400 to > -300 within a few thousand steps.

Dynamics follow the standard Pendulum-v1 definition: state (theta,
theta_dot); obs = [cos(theta), sin(theta), theta_dot]; torque in [-2, 2];
reward = -(angle^2 + 0.1*theta_dot^2 + 0.001*u^2); 200-step episodes.
"""

import numpy as np

from .core import Box, Env


class PendulumEnv(Env):
    MAX_SPEED = 8.0
    MAX_TORQUE = 2.0
    DT = 0.05
    G = 10.0
    M = 1.0
    L = 1.0
    EP_LEN = 200

    def __init__(self, seed: int = 0):
        self.action_space = Box(-self.MAX_TORQUE, self.MAX_TORQUE, (1,))
        self.observation_space = Box(
            np.array([-1.0, -1.0, -self.MAX_SPEED], dtype=np.float32),
            np.array([1.0, 1.0, self.MAX_SPEED], dtype=np.float32),
            (3,))
        self._rng = np.random.default_rng(seed)
        self._th = 0.0
        self._thdot = 0.0
        self._t = 0

    def seed(self, seed: int):
        super().seed(seed)
        self._rng = np.random.default_rng(seed)

    def _obs(self) -> np.ndarray:
        return np.array([np.cos(self._th), np.sin(self._th), self._thdot],
                        dtype=np.float32)

    def reset(self):
        self._th = self._rng.uniform(-np.pi, np.pi)
        self._thdot = self._rng.uniform(-1.0, 1.0)
        self._t = 0
        return self._obs()

    def step(self, action):
        u = float(np.clip(np.asarray(action).reshape(-1)[0],
                          -self.MAX_TORQUE, self.MAX_TORQUE))
        th, thdot = self._th, self._thdot
        angle = ((th + np.pi) % (2 * np.pi)) - np.pi
        cost = angle ** 2 + 0.1 * thdot ** 2 + 0.001 * u ** 2

        thdot_new = thdot + (
            3.0 * self.G / (2.0 * self.L) * np.sin(th)
            + 3.0 / (self.M * self.L ** 2) * u
        ) * self.DT
        thdot_new = float(np.clip(thdot_new, -self.MAX_SPEED, self.MAX_SPEED))
        self._th = th + thdot_new * self.DT
        self._thdot = thdot_new
        self._t += 1
        done = self._t >= self.EP_LEN
        return self._obs(), -float(cost), done, {}
