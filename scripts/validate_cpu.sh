#!/bin/bash
# One-command CPU-side validation (what the driver checks sans GPU):
# build, full non-GPU suite (driver -x mode), bench contract smoke.
set -e
cd "$(dirname "$0")/.."
python -c "import __graft_entry__ as g; g.build()"
python -m pytest tests/ -x -q -m "not gpu"
python -m pytest tests/test_bench_contract.py -q
echo "CPU validation OK"
