"""API-compat shim: reference buffer/replay_buffer.py."""
from torch_actor_critic_amd.buffer.replay import Batch, ReplayBuffer  # noqa: F401
