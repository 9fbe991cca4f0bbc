from .replay import Batch, ReplayBuffer
from .visual import VisualBatch, VisualReplayBuffer

__all__ = ["Batch", "ReplayBuffer", "VisualBatch", "VisualReplayBuffer"]
