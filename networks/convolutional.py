"""API-compat shim: reference networks/convolutional.py."""
from torch_actor_critic_amd.envs.visual import MultiObservation  # noqa: F401
from torch_actor_critic_amd.models.visual import (  # noqa: F401
    VisualActor, VisualCritic, VisualDoubleCritic, calculate_size,
    simple_cnn)
