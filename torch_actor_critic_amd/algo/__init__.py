from .sac import SAC, eval_pi_loss, eval_q_loss, update_targets

__all__ = ["SAC", "eval_pi_loss", "eval_q_loss", "update_targets"]
