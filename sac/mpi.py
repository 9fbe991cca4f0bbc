"""API-compat shim for the reference's MPI layer (reference sac/mpi.py:10-115),
implemented over torch.distributed (RCCL on GPU / gloo on CPU) — the
MI355X replacement for mpi4py.  Function names and contracts match the
reference so downstream code ports unchanged; internally the heavy paths
delegate to torch_actor_critic_amd.parallel.comm (single-bucket
collectives over xGMI).
"""

import numpy as np
import torch

from torch_actor_critic_amd.parallel import comm
from torch_actor_critic_amd.parallel.launch import gpu_fork


def mpi_fork(n: int):
    """Re-launch as n ranks (reference mpi_fork, sac/mpi.py:10-34 — there
    via mpirun re-exec; here via the per-GPU process launcher)."""
    gpu_fork(int(n))
    comm.init_distributed()


def proc_id() -> int:
    return comm.proc_id()


def num_procs() -> int:
    return comm.num_procs()


def allreduce(x, op="sum"):
    t = torch.as_tensor(np.asarray(x), dtype=torch.float64)
    if comm.is_initialized():
        import torch.distributed as dist
        ops = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN,
               "max": dist.ReduceOp.MAX}
        dist.all_reduce(t, op=ops[op])
    return t.numpy()


def mpi_op(x, op):
    scalar = np.isscalar(x)
    arr = np.asarray(x, dtype=np.float64).reshape(-1)
    out = allreduce(arr, op)
    return float(out[0]) if scalar else out


def mpi_sum(x):
    return mpi_op(x, "sum")


def mpi_avg(x):
    return mpi_sum(x) / num_procs()


def setup_pytorch_for_mpi():
    """Divide CPU threads fairly among ranks (reference sac/mpi.py:67-74)."""
    if torch.get_num_threads() == 1:
        return
    fair = max(int(torch.get_num_threads() / num_procs()), 1)
    torch.set_num_threads(fair)


def mpi_avg_grads(module: torch.nn.Module):
    """Average gradients across ranks (reference sac/mpi.py:77-85 does a
    per-tensor Allreduce with NumPy host copies; this does device-side
    collectives — the flat-bucket fast path lives in parallel.comm)."""
    if not comm.is_initialized():
        return
    for p in module.parameters():
        if p.grad is not None:
            comm.allreduce_grads(p.grad)


def broadcast(x, root: int = 0):
    t = torch.as_tensor(x)
    if comm.is_initialized():
        import torch.distributed as dist
        dist.broadcast(t, src=root)
    return t


def sync_params(module: torch.nn.Module):
    """Broadcast parameters from rank 0 (reference sac/mpi.py:93-98)."""
    if not comm.is_initialized():
        return
    for p in module.parameters():
        broadcast(p.data)


def mpi_statistics_scalar(x, with_min_and_max: bool = False):
    """Global mean/std(/min/max) of per-rank lists
    (reference sac/mpi.py:101-115)."""
    x = np.array(x, dtype=np.float32)
    global_sum = mpi_sum(np.sum(x))
    global_n = mpi_sum(len(x))
    mean = global_sum / max(global_n, 1)
    global_sq = mpi_sum(np.sum((x - mean) ** 2))
    std = np.sqrt(global_sq / max(global_n, 1))
    if with_min_and_max:
        gmin = mpi_op(np.min(x) if len(x) else np.inf, "min")
        gmax = mpi_op(np.max(x) if len(x) else -np.inf, "max")
        return mean, std, gmin, gmax
    return mean, std
