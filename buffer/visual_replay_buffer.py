"""API-compat shim: reference buffer/visual_replay_buffer.py."""
from torch_actor_critic_amd.buffer.visual import (  # noqa: F401
    VisualBatch, VisualReplayBuffer)
