"""Pluggable global job scheduler.

Reference: jobserver/driver/JobScheduler.java:22 (SPI: onJobArrival /
onJobFinish / onResourceChange) and SchedulerImpl.java:28-67 (default: run
every arriving job immediately on ALL executors). Select with the
`-scheduler` flag (class name), as the reference's JobServerClient does.
"""

from __future__ import annotations

import importlib
from typing import Dict, List, Optional

from harmony_amd.config import JobConfig


class JobScheduler:
    """SPI. Return the executor ranks a job should run on, or None to queue
    it (re-evaluated when a running job finishes)."""

    def on_job_arrival(self, job: JobConfig, pool: "ResourcePool") -> Optional[List[int]]:
        raise NotImplementedError

    def on_job_finish(self, job_id: str, pool: "ResourcePool") -> None:
        pass

    def on_resource_change(self, pool: "ResourcePool") -> None:
        pass


class ResourcePool:
    """Homogeneous executor pool shared by all jobs (reference
    jobserver/driver/ResourcePool.java:39-106)."""

    def __init__(self, world_size: int):
        self.world_size = world_size
        self.running: Dict[str, List[int]] = {}   # job_id -> ranks

    @property
    def all_ranks(self) -> List[int]:
        return list(range(self.world_size))

    def load(self, rank: int) -> int:
        return sum(1 for ranks in self.running.values() if rank in ranks)


class DefaultScheduler(JobScheduler):
    """Run every arriving job immediately on all executors (reference
    SchedulerImpl default)."""

    def on_job_arrival(self, job, pool):
        return pool.all_ranks


class LeastLoadedScheduler(JobScheduler):
    """Place each job on the `num_executors` least-loaded executors
    (app_args key 'num_executors', default all)."""

    def on_job_arrival(self, job, pool):
        n = int(job.app_args.get("num_executors", pool.world_size))
        n = max(1, min(n, pool.world_size))
        ranks = sorted(pool.all_ranks, key=lambda r: (pool.load(r), r))[:n]
        return sorted(ranks)


class FIFOExclusiveScheduler(JobScheduler):
    """One job at a time on all executors; later jobs queue."""

    def on_job_arrival(self, job, pool):
        if pool.running:
            return None
        return pool.all_ranks


_BUILTIN = {
    "default": DefaultScheduler,
    "least_loaded": LeastLoadedScheduler,
    "fifo": FIFOExclusiveScheduler,
}


def load_scheduler(name: str) -> JobScheduler:
    """Builtin name or 'module.path:ClassName'."""
    if name in _BUILTIN:
        return _BUILTIN[name]()
    if ":" in name:
        mod, cls = name.split(":", 1)
        return getattr(importlib.import_module(mod), cls)()
    raise KeyError(f"unknown scheduler '{name}' (builtins: {sorted(_BUILTIN)})")
