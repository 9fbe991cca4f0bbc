"""HIP/CDNA4 kernel extension loading and dispatch.

The native extension (``_tac_hip``) is built in-tree from
``torch_actor_critic_amd/ops/csrc`` for gfx950 only (no CUDA shim, no
dual path).  On a GPU machine the fused ops MUST come from the extension:
if a CUDA(HIP) tensor reaches an op and the extension is missing we raise
loudly rather than silently falling back to eager PyTorch.  On CPU (the
plumbing / unit-test path, reference BASELINE config 1) the ops fall back
to eager PyTorch implementations that define the numerics contract.
"""

import os

import torch

_EXT = None
_EXT_ERR = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib
        _EXT = importlib.import_module("torch_actor_critic_amd.ops._tac_hip")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e
        _EXT = None
    return _EXT


def extension():
    """Return the loaded HIP extension module or None (CPU-only mode)."""
    return _try_load()


def has_extension() -> bool:
    return _try_load() is not None


def require_extension():
    """Fetch the extension, raising loudly if a GPU is present but the
    native kernels were not built (the framework must never silently run
    eager PyTorch on the GPU hot path)."""
    ext = _try_load()
    if ext is None:
        raise RuntimeError(
            "torch_actor_critic_amd HIP extension (_tac_hip) is not built. "
            "Run `python -m torch_actor_critic_amd.ops.build` (hipcc, "
            "--offload-arch=gfx950) before using GPU tensors. "
            f"Original import error: {_EXT_ERR!r}"
        )
    return ext


def use_native(*tensors) -> bool:
    """True if these tensors should run the hand-written HIP path.

    Any CUDA (ROCm) tensor routes to the native kernels; if the extension
    is absent that is a hard error.  Setting TAC_AMD_FORCE_EAGER=1
    bypasses the kernels (debug only — numerics A/B).
    """
    if os.environ.get("TAC_AMD_FORCE_EAGER") == "1":
        return False
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    require_extension()
    return True
