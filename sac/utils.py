"""API-compat shim for the reference's sac/utils.py normalizers."""
from torch_actor_critic_amd.utils.normalizer import (  # noqa: F401
    Identity, StateNormalizer, WelfordVarianceEstimate)
