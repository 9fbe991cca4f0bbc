from .comm import (allreduce_grads, allreduce_mean, barrier, gather_stats,
                   init_distributed, is_initialized, num_procs, proc_id,
                   statistics_scalar, sync_flat_params)
from .flat import FlatParams, flatten_module_like
from .launch import gpu_fork, in_worker

__all__ = [
    "init_distributed", "is_initialized", "proc_id", "num_procs", "barrier",
    "sync_flat_params", "allreduce_grads", "allreduce_mean", "gather_stats",
    "statistics_scalar", "FlatParams", "flatten_module_like", "gpu_fork",
    "in_worker",
]
