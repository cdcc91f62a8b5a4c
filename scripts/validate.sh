#!/usr/bin/env bash
# One-stop validation: build + CPU suite (+ GPU suite & bench on a GPU box).
set -e
cd "$(dirname "$0")/.."
echo "== build (gfx950 cross-compile) =="
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
echo "== CPU suite =="
python -m pytest tests -q -m "not gpu"
if python -c "import torch; raise SystemExit(0 if torch.cuda.is_available() else 1)"; then
  echo "== GPU suite =="
  python -m pytest tests -q -m gpu
  echo "== smoke =="
  python -c "import __graft_entry__ as g; g.smoke()"
  echo "== bench (short) =="
  python bench.py --steps 10 --warmup 3
fi
echo "ALL GREEN"
