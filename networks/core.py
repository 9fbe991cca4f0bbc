"""API-compat shim: reference networks/core.py."""
from torch_actor_critic_amd.models.mlp import mlp  # noqa: F401
