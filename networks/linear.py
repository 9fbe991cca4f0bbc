"""API-compat shim: reference networks/linear.py."""
from torch_actor_critic_amd.models.mlp import (  # noqa: F401
    Actor, Critic, DoubleCritic)
