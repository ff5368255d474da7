#!/bin/bash
# Submit a nmf job to the running job server (reference submit_nmf.sh flags).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client submit -app nmf "$@"
