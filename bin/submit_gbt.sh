#!/bin/bash
# Submit a gbt job to the running job server (reference submit_gbt.sh flags).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client submit -app gbt "$@"
