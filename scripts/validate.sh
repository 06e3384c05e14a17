#!/usr/bin/env bash
# Full local (CPU) validation ladder — what the driver checks, runnable
# anywhere: style gate, gfx950 cross-build, CPU test suite.
# GPU-side: pytest -m gpu, __graft_entry__.smoke(), bench.py (run on a
# MI355X box).
set -e
cd "$(dirname "$0")/.."
echo "== style =="
python tools/style_check.py
echo "== build (gfx950 cross-compile) =="
python -c "import __graft_entry__ as g; g.build()"
echo "== CPU suite =="
python -m pytest tests -q -m "not gpu"
echo "ALL LOCAL CHECKS PASSED"
