#!/bin/bash
# Submit a mlr job to the running job server (reference submit_mlr.sh flags).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client submit -app mlr "$@"
