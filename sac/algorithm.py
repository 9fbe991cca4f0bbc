"""API-compat shim: the reference's `sac.algorithm` surface
(reference sac/algorithm.py) backed by the MI355X-native implementation."""
from torch_actor_critic_amd.algo.sac import (  # noqa: F401
    SAC, eval_pi_loss, eval_q_loss, update_targets)
