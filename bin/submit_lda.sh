#!/bin/bash
# Submit a lda job to the running job server (reference submit_lda.sh flags).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client submit -app lda "$@"
