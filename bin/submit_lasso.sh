#!/bin/bash
# Submit a lasso job to the running job server (reference submit_lasso.sh flags).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client submit -app lasso "$@"
