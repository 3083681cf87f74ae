#!/usr/bin/env bash
# One-command verification: build + CPU suite (+ GPU suite when a GPU is
# present). Mirrors the round driver's checks.
set -e
cd "$(dirname "$0")/.."
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
python -m pytest tests/ -x -q -m "not gpu"
if python -c "import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests/ -x -q -m gpu
  python __graft_entry__.py smoke
fi
echo "ALL CHECKS PASSED"
