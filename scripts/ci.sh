#!/bin/bash
# CI-lite: what the driver runs on CPU each round, in one command.
#   bash scripts/ci.sh
set -e
cd "$(dirname "$0")/.."
echo "== build (gfx950 cross-compile) =="
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
echo "== CPU test suite =="
python -m pytest tests/ -x -q -m "not gpu"
echo "== entrypoint check =="
python -c "import __graft_entry__; print('graft entry importable')"
echo "CI OK"
