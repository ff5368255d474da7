#!/bin/bash
cd "$(dirname "$0")/.."
exec python -m harmony_amd.jobserver.client submit -app pagerank "$@"
