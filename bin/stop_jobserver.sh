#!/bin/bash
# Shut down the job server (waits for running jobs) — reference stop_jobserver.sh.
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client shutdown "$@"
