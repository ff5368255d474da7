#!/bin/bash
# Standalone shortest_path run without the job server (reference run_shortest_path.sh /
# ETDolphinLauncher mode). Multi-GPU: wrap with torchrun (see
# harmony_amd/standalone.py).
cd "$(dirname "$0")/.."
exec python -m harmony_amd.standalone -app shortest_path "$@"
