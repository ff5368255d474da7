#!/bin/bash
# End-to-end breadth showcase on one MI355X: job server via CLI scripts,
# all five PS apps + both Pregel apps, elastic reconfiguration, offline
# model eval. Output log is committed under profiles/ as round evidence.
set -x
mkdir -p gpurun_out
( MASTER_PORT=29650 timeout 560 bin/start_jobserver.sh -num_executors 1 -port 7311 > gpurun_out/showcase_server.log 2>&1 & )
sleep 25
T=90
timeout $T bin/submit_nmf.sh -port 7311 -job_id sc_nmf -max_num_epochs 3 -num_mini_batches 4 -num_cols 16384 -rank 100 -rows_per_batch 4096 -nnz_per_row 64 --wait | head -c 400
echo
timeout $T bin/submit_mlr.sh -port 7311 -job_id sc_mlr -max_num_epochs 3 -num_mini_batches 4 -num_classes 10 -num_features 4096 -num_parts_per_class 8 -batch_size 4096 -model_chkp_per_epoch true -offline_model_eval true --wait | head -c 600
echo
timeout $T bin/submit_lda.sh -port 7311 -job_id sc_lda -max_num_epochs 2 -num_mini_batches 4 -num_vocabs 50000 -num_topics 1024 -docs_per_batch 4096 -tokens_per_doc 64 -sampler alias --wait | head -c 400
echo
timeout $T bin/submit_gbt.sh -port 7311 -job_id sc_gbt -max_num_epochs 4 -num_mini_batches 2 -num_features 32 -batch_size 8192 -num_bins 64 -max_depth 5 --wait | head -c 400
echo
timeout $T bin/submit_lasso.sh -port 7311 -job_id sc_lasso -max_num_epochs 3 -num_mini_batches 2 -num_features 256 -num_parts 16 -batch_size 4096 --wait | head -c 400
echo
timeout $T bin/submit_pagerank.sh -port 7311 -job_id sc_pr -num_vertices 100000 -out_degree 8 -num_iters 20 --wait | head -c 400
echo
timeout $T bin/submit_shortest_path.sh -port 7311 -job_id sc_sp -num_vertices 100000 -out_degree 4 --wait | head -c 400
echo
# async one-sided mode (hipIpc/xGMI data plane, no collectives)
timeout $T bin/submit_mlr.sh -port 7311 -job_id sc_async -max_num_epochs 2 -num_mini_batches 4 -num_classes 5 -num_features 2048 -num_parts_per_class 8 -batch_size 2048 -step_size 0.5 -one_sided true --wait | head -c 300
echo
# restore-from-checkpoint (reference createTable(chkpId))
timeout $T bin/submit_mlr.sh -port 7311 -job_id sc_restore -max_num_epochs 1 -num_mini_batches 4 -num_classes 10 -num_features 4096 -num_parts_per_class 8 -batch_size 4096 -restore_chkp sc_mlr/epoch2 --wait | head -c 300
echo
# elastic: addvector with the scripted rotating optimizer + validation
timeout $T python -m harmony_amd.jobserver.client submit -port 7311 -app addvector -job_id sc_elastic -max_num_epochs 4 -num_mini_batches 6 -optimizer homogeneous -optimizer_period 4 --wait 2>/dev/null | head -c 200
echo
bin/stop_jobserver.sh -port 7311 | head -c 100
echo
