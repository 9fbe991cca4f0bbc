"""Tracing / profiling helpers.

The reference has no tracing at all.  Here every hot
phase of the training loop can be bracketed with roctx ranges so
``rocprofv3 --kernel-trace`` / ``--sys-trace`` attributes kernels to
algorithm phases, plus lightweight wall-clock throughput counters
(updates/sec, env-steps/sec — the BASELINE.json metric) that SAC.train
reports per epoch.
"""

import contextlib
import ctypes
import time


_roctx = None
_roctx_tried = False


def _load_roctx():
    global _roctx, _roctx_tried
    if _roctx_tried:
        return _roctx
    _roctx_tried = True
    for name in ("libroctx64.so", "libroctx64.so.4", "librocprofiler-sdk-roctx.so"):
        try:
            _roctx = ctypes.CDLL(name)
            _roctx.roctxRangePushA.argtypes = [ctypes.c_char_p]
            break
        except OSError:
            continue
    return _roctx


@contextlib.contextmanager
def roctx_range(name: str):
    """Bracket a phase with a roctx range (no-op when roctx is absent)."""
    lib = _load_roctx()
    if lib is not None:
        lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        if lib is not None:
            lib.roctxRangePop()


class Throughput:
    """Rolling updates/sec + env-steps/sec counters."""

    def __init__(self):
        self.reset()

    def reset(self):
        self._t0 = time.perf_counter()
        self.env_steps = 0
        self.updates = 0

    def tick_env(self, n: int = 1):
        self.env_steps += n

    def tick_update(self, n: int = 1):
        self.updates += n

    def rates(self):
        dt = max(time.perf_counter() - self._t0, 1e-9)
        return {"env_steps_per_sec": self.env_steps / dt,
                "updates_per_sec": self.updates / dt}
