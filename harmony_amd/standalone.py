"""Standalone (non-jobserver) job launcher — reference ETDolphinLauncher
mode (dolphin/core/client/ETDolphinLauncher.java:111,219: `run_*.sh`
allocate their own executors and run one job to completion).

MI355X form: torchrun spawns one process per GPU; each rank runs the job
directly through run_job (no job server, no scheduler). Single-process
invocation works too (world 1):

  bin/run_nmf.sh -max_num_epochs 5 -num_mini_batches 4 -rank 100 ...
  # or N GPUs:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 -m harmony_amd.standalone -app nmf ...

Flags use the reference's `-name value` convention (same parser as the
jobserver client); JobConfig fields split from app args automatically.
"""

from __future__ import annotations

import json
import sys
import uuid


def main() -> None:
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.jobserver.client import _parse_flags
    from harmony_amd.pregel.runner import PREGEL_APPS, run_pregel_job
    from harmony_amd.runtime.bootstrap import init_executor

    job_kw, app_args, _wait = _parse_flags(sys.argv[1:])
    app = job_kw.pop("app", None)
    if not app:
        raise SystemExit("standalone mode requires -app <name>")
    device = job_kw.pop("device", "auto")
    job_kw.pop("port", None)       # jobserver-only flags are ignored
    job_kw.pop("timeout", None)
    job_id = job_kw.pop("job_id", f"{app}-{uuid.uuid4().hex[:8]}")
    job = JobConfig(job_id=job_id, app=app, app_args=app_args, **job_kw)
    ctx = init_executor(RuntimeConfig(device=device))
    if app in PREGEL_APPS:
        from harmony_amd.jobserver.server import JobView

        view = JobView(rank=ctx.rank, world_size=ctx.world_size,
                       device=ctx.device, store=ctx.store, group=None,
                       global_ranks=list(range(ctx.world_size)))
        summary = run_pregel_job(job, view)
    else:
        summary = run_job(job, ctx).summary()
    if ctx.rank == 0:
        print(json.dumps(summary))


if __name__ == "__main__":
    main()
