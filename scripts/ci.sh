#!/usr/bin/env bash
# CI entry (the reference's .travis.yml analog): build the gfx950 extension,
# run the CPU test tier, and — when a GPU is visible — the GPU tier too.
set -euo pipefail
cd "$(dirname "$0")/.."
python build_native.py
python -m pytest tests -x -q -m "not gpu"
if python -c "import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests -x -q -m gpu
fi
