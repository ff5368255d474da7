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

    def order(self, pending: List[JobConfig]) -> List[JobConfig]:
        """Order in which queued jobs are offered resources (default FIFO)."""
        return pending


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


class PriorityScheduler(JobScheduler):
    """Exclusive one-job-at-a-time like fifo, but the queue drains in
    priority order: higher `-priority` first (app_args, default 0), FIFO
    within a priority level."""

    def on_job_arrival(self, job, pool):
        if pool.running:
            return None
        return pool.all_ranks

    def order(self, pending):
        return sorted(pending,
                      key=lambda j: -int(j.app_args.get("priority", 0)))


class GangScheduler(JobScheduler):
    """Each job gets `num_executors` executors EXCLUSIVELY (no co-location);
    queues until that many executors are idle. The gang analogue for jobs
    that cannot share GPUs."""

    def on_job_arrival(self, job, pool):
        n = int(job.app_args.get("num_executors", pool.world_size))
        n = max(1, min(n, pool.world_size))
        free = [r for r in pool.all_ranks if pool.load(r) == 0]
        if len(free) < n:
            return None
        return free[:n]


_BUILTIN = {
    "default": DefaultScheduler,
    "least_loaded": LeastLoadedScheduler,
    "fifo": FIFOExclusiveScheduler,
    "priority": PriorityScheduler,
    "gang": GangScheduler,
}


def load_scheduler(name: str) -> JobScheduler:
    """Builtin name or 'module.path:ClassName'."""
    if name in _BUILTIN:
        return _BUILTIN[name]()
    if ":" in name:
        mod, cls = name.split(":", 1)
        return getattr(importlib.import_module(mod), cls)()
    raise KeyError(f"unknown scheduler '{name}' (builtins: {sorted(_BUILTIN)})")
