"""Environment registry — explicit registration (fixes the reference's
import-side-effect registration, SURVEY.md Q10 / environments/__init__.py:4-7)."""

from .core import Box, Env, make, register, registered
from .pendulum import PendulumEnv
from .synthetic import MUJOCO_SHAPES, SyntheticMujocoEnv, make_factory
from .visual import (DeepMindWallRunner, MultiObservation, VisualCheetahRun)

register("Pendulum-v1", PendulumEnv)
register("Pendulum-v0", PendulumEnv)
for _base in MUJOCO_SHAPES:
    for _ver in ("v2", "v3", "v4"):
        register(f"{_base}-{_ver}", make_factory(_base))
register("DeepMindWallRunner-v0", DeepMindWallRunner)
register("VisualCheetahRun-v0", VisualCheetahRun)

__all__ = [
    "Box", "Env", "make", "register", "registered", "PendulumEnv",
    "SyntheticMujocoEnv", "MultiObservation", "DeepMindWallRunner",
    "VisualCheetahRun", "MUJOCO_SHAPES",
]
