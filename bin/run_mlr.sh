#!/bin/bash
# Standalone mlr run without the job server (reference run_mlr.sh /
# ETDolphinLauncher mode). Multi-GPU: wrap with torchrun (see
# harmony_amd/standalone.py).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.standalone -app mlr "$@"
