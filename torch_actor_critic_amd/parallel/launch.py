"""Single-node multi-GPU launcher.

Replaces the reference's ``mpi_fork`` self-re-exec under mpirun
(``sac/mpi.py:10-34``) while keeping its one-flag UX: ``gpu_fork(n)``
re-launches ``sys.argv`` as n processes, one per GPU (or n CPU ranks when
no GPU is present), with torchrun-style env vars, then the parent waits
and exits.  Unlike the reference it is called BEFORE any
heavy state is built, and the flag is typed.
"""

import os
import subprocess
import sys


def in_worker() -> bool:
    return os.environ.get("TAC_AMD_WORKER") == "1" or \
        int(os.environ.get("WORLD_SIZE", "1")) > 1


def gpu_fork(n: int, master_port: int = 29511):
    """Re-launch the current script as n ranks; parent waits and exits.
    No-op when n<=1 or when we already are a worker."""
    n = int(n)
    if n <= 1 or in_worker():
        return
    procs = []
    for rank in range(n):
        env = dict(os.environ)
        env.update(
            TAC_AMD_WORKER="1",
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE=str(n),
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(master_port),
        )
        procs.append(subprocess.Popen([sys.executable] + sys.argv, env=env))
    code = 0
    try:
        for p in procs:
            p.wait()
            code = code or p.returncode
    except KeyboardInterrupt:
        for p in procs:
            p.terminate()
        code = 130
    sys.exit(code)
