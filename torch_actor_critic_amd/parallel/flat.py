"""Flat contiguous parameter/gradient storage.

The reference iterates per-parameter for every maintenance op: polyak is
a Python loop of mul_/add_ pairs (sac/algorithm.py:77-81), gradient
averaging is one MPI Allreduce per tensor with host round-trips
(sac/mpi.py:77-85), and Adam is torch.optim's per-tensor loop.  On
MI355X these ops are latency-bound (the whole actor is ~0.3 MB), so this
framework flattens every module's parameters into ONE contiguous HBM
buffer at construction:

* polyak target update  -> ONE axpby kernel over the flat buffer
* Adam                  -> ONE fused kernel over flat (p, g, m, v)
* DP gradient reduction -> ONE RCCL all-reduce of the flat grad bucket
  (xGMI is per-link latency-bound at sub-MB payloads: 1 call, not 20)
* initial weight sync   -> ONE RCCL broadcast of the flat param buffer

Parameter tensors stay real ``nn.Parameter`` views into the flat buffer,
so ``state_dict()`` / checkpoint layout / optimizer APIs are unchanged.
"""

import typing as t

import torch
import torch.nn as nn


class FlatParams:
    """Flatten a module's parameters (and gradients) into contiguous
    device buffers, re-pointing each parameter/grad at a view."""

    def __init__(self, module: nn.Module):
        self.module = module
        params = [p for p in module.parameters() if p.requires_grad]
        if not params:
            raise ValueError("module has no trainable parameters")
        dev = params[0].device
        dtype = params[0].dtype
        total = sum(p.numel() for p in params)
        self.flat = torch.zeros(total, dtype=dtype, device=dev)
        self.flat_grad = torch.zeros(total, dtype=dtype, device=dev)
        self.numel = total
        self._params = params
        self._slices: t.List[t.Tuple[int, int]] = []
        off = 0
        for p in params:
            n = p.numel()
            self.flat[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat[off:off + n].view_as(p.data)
            p.grad = self.flat_grad[off:off + n].view_as(p.data)
            self._slices.append((off, n))
            off += n

    def zero_grad(self):
        self.flat_grad.zero_()

    def check_views(self) -> bool:
        """True iff autograd still accumulates into our flat grad buffer
        (a replaced .grad tensor would silently break the single-bucket
        all-reduce — asserted by tests)."""
        base = self.flat_grad.data_ptr()
        for p, (off, n) in zip(self._params, self._slices):
            if p.grad is None:
                return False
            if p.grad.data_ptr() != base + off * self.flat_grad.element_size():
                return False
            if p.data.data_ptr() != (self.flat.data_ptr()
                                     + off * self.flat.element_size()):
                return False
        return True

    def params(self):
        return self._params


def flatten_module_like(module: nn.Module) -> torch.Tensor:
    """One contiguous copy of a module's parameter data (for frozen
    target networks that need only the flat view, no grads)."""
    with torch.no_grad():
        params = [p for p in module.parameters()]
        total = sum(p.numel() for p in params)
        flat = torch.zeros(total, dtype=params[0].dtype,
                           device=params[0].device)
        off = 0
        for p in params:
            n = p.numel()
            flat[off:off + n].copy_(p.data.reshape(-1))
            p.data = flat[off:off + n].view_as(p.data)
            off += n
        return flat
