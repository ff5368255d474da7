#!/bin/bash
# Start the long-running job server: one executor process per GPU.
# Usage: ./start_jobserver.sh [-num_executors N] [-scheduler NAME] [-port P] [-device auto|cpu|cuda]
# (reference: jobserver/bin/start_jobserver.sh — flags -num_executors, -scheduler)
N=$(python3 -c "import torch; print(torch.cuda.device_count() or 1)" 2>/dev/null || echo 1)
SCHED=default; PORT=7008; DEVICE=auto; EXTRA=()
while [ $# -gt 0 ]; do
  case "$1" in
    -num_executors|--num_executors) N=$2; shift 2;;
    -scheduler|--scheduler) SCHED=$2; shift 2;;
    -port|--port) PORT=$2; shift 2;;
    -device|--device) DEVICE=$2; shift 2;;
    *) EXTRA+=("$1"); shift;;
  esac
done
cd "$(dirname "$0")/.."
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
  --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
  -m harmony_amd.jobserver.server -scheduler "$SCHED" -port "$PORT" -device "$DEVICE" "${EXTRA[@]}"
