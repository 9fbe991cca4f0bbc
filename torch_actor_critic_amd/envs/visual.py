"""Visual environments: MultiObservation (features + frame) observations.

``MultiObservation`` matches the reference's dataclass
(reference environments/wall_runner.py:11-14) consumed by the visual
networks and the visual replay buffer.  ``DeepMindWallRunner`` is the
wall-runner surrogate: same spaces (act Box(-1,1,(56,)), obs 168-dim
features + 3x64x64 egocentric frame — reference wall_runner.py:20-21,
38-58) on synthetic dynamics (dm_control is not installable here).
``VisualCheetahRun`` covers BASELINE config 4 (cheetah-run from pixels,
3x84x84).
"""

from dataclasses import dataclass

import numpy as np
import torch

from .core import Box, Env
from .synthetic import SyntheticMujocoEnv


@dataclass
class MultiObservation:
    features: torch.FloatTensor
    frame: torch.FloatTensor


class _SyntheticVisualEnv(Env):
    """Wraps a synthetic feature-space env and renders a cheap procedural
    frame from its state (a few state-dependent gradients/bars), so the
    conv encoder has state-correlated pixels to learn from."""

    def __init__(self, feat_dim: int, act_dim: int, vis_dim, act_limit=1.0,
                 ep_len: int = 1000, seed: int = 0):
        self.vis_dim = tuple(vis_dim)  # (C, H, W)
        self._inner = SyntheticMujocoEnv(feat_dim, act_dim, act_limit,
                                         ep_len, seed)
        self.action_space = self._inner.action_space
        self.observation_space = Box(-1, 1, (feat_dim,))
        c, h, w = self.vis_dim
        rng = np.random.default_rng(777)
        # fixed projection state -> per-row intensity pattern
        self._proj = (rng.standard_normal((h, feat_dim)) / np.sqrt(feat_dim)
                      ).astype(np.float32)
        self._cols = np.linspace(-1, 1, w, dtype=np.float32)

    def seed(self, seed: int):
        self._inner.seed(seed)

    def _render_frame(self, feats: np.ndarray) -> np.ndarray:
        c, h, w = self.vis_dim
        rows = np.tanh(self._proj @ feats)               # (h,)
        img = rows[:, None] * self._cols[None, :]        # (h, w)
        frame = np.broadcast_to(img, (c, h, w)).copy()
        return frame.astype(np.float32)

    def _obs(self, feats: np.ndarray) -> MultiObservation:
        return MultiObservation(
            torch.from_numpy(np.ascontiguousarray(feats, dtype=np.float32)),
            torch.from_numpy(self._render_frame(feats)),
        )

    def process_observations(self, feats) -> MultiObservation:
        """Build a MultiObservation from raw per-step features (the
        reference's method name, environments/wall_runner.py:38-58 —
        there it concatenates dm_control sensor groups and rolls the
        egocentric camera; here the synthetic dynamics provide the
        feature vector directly)."""
        return self._obs(np.asarray(feats, dtype=np.float32))

    def reset(self):
        return self._obs(self._inner.reset())

    def step(self, action):
        feats, reward, done, info = self._inner.step(action)
        return self._obs(feats), reward, done, info


class DeepMindWallRunner(_SyntheticVisualEnv):
    """Wall-runner surrogate: 168 features, 3x64x64 frame, 56-dim action
    (reference environments/wall_runner.py:17-21)."""

    def __init__(self):
        super().__init__(feat_dim=168, act_dim=56, vis_dim=(3, 64, 64),
                         act_limit=1.0, ep_len=1000)


class VisualCheetahRun(_SyntheticVisualEnv):
    """dm_control cheetah-run-from-pixels surrogate: 17 features,
    3x84x84 frame, 6-dim action (BASELINE config 4)."""

    def __init__(self):
        super().__init__(feat_dim=17, act_dim=6, vis_dim=(3, 84, 84),
                         act_limit=1.0, ep_len=1000)
