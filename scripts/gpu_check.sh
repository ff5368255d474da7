#!/bin/bash
# Round-1 first GPU validation: numerics tests, smoke, short bench, rocprof.
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
rocm-smi --showproductname 2>/dev/null | head -5 > gpurun_out/gpu.txt
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
timeout 600 python -c 'import __graft_entry__; __graft_entry__.smoke()' > gpurun_out/smoke.log 2>&1
echo "smoke exit: $?" >> gpurun_out/smoke.log
timeout 900 python bench.py --steps 10 --warmup 3 > gpurun_out/bench1.log 2>&1
echo "bench exit: $?" >> gpurun_out/bench1.log
cd /tmp && export TMPDIR=/tmp
timeout 900 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof -- python $R/bench.py --steps 3 --warmup 1 > $R/gpurun_out/rocprof.log 2>&1
echo "rocprof exit: $?" >> $R/gpurun_out/rocprof.log
tail -5 $R/gpurun_out/pytest_gpu.log $R/gpurun_out/smoke.log $R/gpurun_out/bench1.log
