"""API-compat shim: reference environments/wall_runner.py."""
from torch_actor_critic_amd.envs.visual import (  # noqa: F401
    DeepMindWallRunner, MultiObservation)
