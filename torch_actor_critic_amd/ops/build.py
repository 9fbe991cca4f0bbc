"""In-tree build of the gfx950 HIP extension.

Builds ``_tac_hip`` from csrc/ with hipcc (gfx950 via PYTORCH_ROCM_ARCH —
cross-compiles fine on a machine with no GPU) and copies the .so next to
this file so it imports as ``torch_actor_critic_amd.ops._tac_hip`` and
travels with the repo snapshot to GPU machines (no JIT-cache dependency).

Run:  python -m torch_actor_critic_amd.ops.build
"""

import glob
import os
import shutil
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_OPS_DIR, "csrc")
_BUILD = os.path.join(_OPS_DIR, "_build")


def build(verbose: bool = False) -> str:
    os.environ["PYTORCH_ROCM_ARCH"] = "gfx950"
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils import cpp_extension

    os.makedirs(_BUILD, exist_ok=True)
    sources = sorted(glob.glob(os.path.join(_CSRC, "*.hip")))
    assert sources, "no HIP sources found"

    cpp_extension.load(
        name="_tac_hip",
        sources=sources,
        build_directory=_BUILD,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3"],
        verbose=verbose,
        is_python_module=False,
    )
    cands = glob.glob(os.path.join(_BUILD, "_tac_hip*.so"))
    assert cands, "built extension .so not found"
    so_dst = os.path.join(_OPS_DIR, "_tac_hip.so")
    shutil.copy2(cands[0], so_dst)
    return so_dst


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv)
    print(f"built {path}")
