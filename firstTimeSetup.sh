#!/usr/bin/env bash
# First-time setup (MI355X / ROCm counterpart of the reference's
# firstTimeSetup.sh): build the gfx950 HIP extension in-tree and install
# the dev hooks.  Requires ROCm (hipcc) and PyTorch-ROCm on PATH; no
# network access is needed beyond what pip already has cached.
set -euo pipefail
cd "$(dirname "$0")"

echo "== building the gfx950 HIP extension (in-tree .so) =="
python -m torch_actor_critic_amd.ops.build

echo "== sanity: import + CPU test suite =="
python -c "import torch_actor_critic_amd; print('import OK')"
python -m pytest tests -q -m "not gpu"

if command -v pre-commit >/dev/null 2>&1; then
  echo "== installing pre-commit hooks =="
  pre-commit install
else
  echo "(pre-commit not installed; skipping hook setup)"
fi

echo "Setup complete.  GPU suite: python -m pytest tests -q -m gpu"
