#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu.log
timeout 600 python bench.py --steps 10 --warmup 3 > gpurun_out/bench_3job.log 2>&1
for app in nmf mlr lda; do
  timeout 420 python bench.py --apps $app --steps 10 --warmup 3 > gpurun_out/bench_$app.log 2>&1
done
timeout 420 python bench.py --apps lda --steps 6 --warmup 2 --lda-topics 1024 > gpurun_out/bench_lda_k1024.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof2 -- python $R/bench.py --steps 3 --warmup 1 > $R/gpurun_out/rocprof2.log 2>&1
tail -2 $R/gpurun_out/pytest_gpu.log; grep -h '"value"' $R/gpurun_out/bench_*.log | head -8
