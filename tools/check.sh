#!/bin/sh
# Developer gate: CPU test tier + gfx950 cross-compile (mirrors what
# the CI driver runs on a GPU-less host).
set -e
cd "$(dirname "$0")/.."
python -m pytest tests/ -x -q -m "not gpu"
python -c "import __graft_entry__ as g; g.build()"
echo "check OK"
