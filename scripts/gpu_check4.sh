#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu4.log 2>&1
echo "pytest exit: $?" >> gpurun_out/pytest_gpu4.log
timeout 420 python bench.py --steps 10 --warmup 3 > gpurun_out/bench4_3job.log 2>&1
for app in nmf mlr lda; do
  timeout 300 python bench.py --apps $app --steps 10 --warmup 3 > gpurun_out/bench4_$app.log 2>&1
done
# jobserver end-to-end on GPU via the CLI scripts
( MASTER_PORT=29610 timeout 300 bin/start_jobserver.sh -num_executors 1 -port 7206 > gpurun_out/js_server.log 2>&1 & )
sleep 25
timeout 120 bin/submit_nmf.sh -port 7206 -job_id jsg_nmf -max_num_epochs 2 -num_mini_batches 2 -num_cols 4096 -rank 64 -rows_per_batch 1024 -nnz_per_row 32 --wait > gpurun_out/js_nmf.log 2>&1
timeout 120 bin/submit_lda.sh -port 7206 -job_id jsg_lda -max_num_epochs 2 -num_mini_batches 2 -num_vocabs 5000 -num_topics 64 -docs_per_batch 512 -tokens_per_doc 32 --wait > gpurun_out/js_lda.log 2>&1
timeout 120 bin/submit_pagerank.sh -port 7206 -job_id jsg_pr -num_vertices 4096 -num_iters 10 --wait > gpurun_out/js_pr.log 2>&1
timeout 60 bin/stop_jobserver.sh -port 7206 > gpurun_out/js_stop.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof4 -- python $R/bench.py --steps 3 --warmup 1 > $R/gpurun_out/rocprof4.log 2>&1
tail -2 $R/gpurun_out/pytest_gpu4.log
grep -ho '"ms_per_step": [0-9.]*' $R/gpurun_out/bench4_*.log
head -c 300 $R/gpurun_out/js_nmf.log; echo; head -c 200 $R/gpurun_out/js_pr.log
