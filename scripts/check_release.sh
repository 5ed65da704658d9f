#!/bin/bash
# One-command verification mirroring the round driver's checks.
#   CPU box:  bash scripts/check_release.sh
#   GPU box:  bash scripts/check_release.sh gpu
set -e
cd "$(dirname "$0")/.."
echo "== build (hipcc gfx950, in-tree) =="
python -c 'import __graft_entry__; __graft_entry__.build()'
echo "== CPU test suite =="
python -m pytest tests -q -m "not gpu"
if [ "$1" = "gpu" ]; then
  echo "== GPU test suite =="
  python -m pytest tests -q -m gpu
  echo "== smoke =="
  python -c 'import __graft_entry__; __graft_entry__.smoke()'
  echo "== bench (default contract) =="
  python bench.py
fi
echo "ALL CHECKS PASSED"
