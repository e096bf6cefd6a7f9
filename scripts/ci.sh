#!/usr/bin/env bash
# CI entry (the reference's .travis.yml analog): build the gfx950 extension,
# run the CPU test tier, the host ASan/UBSan tier, and — when a GPU is
# visible — the GPU tier too. TFREC_SKIP_SANITIZE=1 skips the ASan pass.
set -euo pipefail
cd "$(dirname "$0")/.."

python build_native.py
python -m pytest tests -x -q -m "not gpu"

if [ "${TFREC_SKIP_SANITIZE:-0}" != "1" ]; then
  echo "== sanitizer tier (host ASan/UBSan build) =="
  python build_native.py --force --sanitize
  ASAN_RT=$(hipcc -print-file-name=libclang_rt.asan-x86_64.so)
  # leak detection off: CPython itself holds intentional leaks; the dist
  # tests are multiprocess-spawn and covered unsanitized above
  LD_PRELOAD="$ASAN_RT" ASAN_OPTIONS=detect_leaks=0 \
    python -m pytest tests -x -q -m "not gpu" --deselect tests/test_dist.py
  python build_native.py --force   # restore the regular build
fi

if python -c "import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)"; then
  python -m pytest tests -x -q -m gpu
fi
