from . import checkpoint
from .normalizer import Identity, StateNormalizer, WelfordVarianceEstimate

__all__ = ["checkpoint", "StateNormalizer", "WelfordVarianceEstimate",
           "Identity"]
