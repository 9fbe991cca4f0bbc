from .mlp import Actor, Critic, DoubleCritic, mlp
from .visual import (VisualActor, VisualCritic, VisualDoubleCritic,
                     calculate_size, simple_cnn)

__all__ = [
    "mlp", "Actor", "Critic", "DoubleCritic", "calculate_size",
    "simple_cnn", "VisualActor", "VisualCritic", "VisualDoubleCritic",
]
