"""API-compat shim: the reference registered DeepMindWallRunner-v0 on
import (reference environments/__init__.py:4-7); here registration
happens explicitly in torch_actor_critic_amd.envs and
importing this package simply ensures it ran."""
import torch_actor_critic_amd.envs  # noqa: F401
