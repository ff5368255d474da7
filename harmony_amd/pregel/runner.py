"""Pregel job runner — jobserver glue (reference pregel/jobserver/
PregelJobEntity + PregelMaster.start)."""

from __future__ import annotations

import time
from typing import Optional

from harmony_amd.config import JobConfig
from harmony_amd.pregel.engine import PregelEngine
from harmony_amd.pregel.graphapps import (PageRankComputation,
                                          ShortestPathComputation,
                                          make_ring_plus_random_graph)
from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler
from harmony_amd.utils import stable_seed

PREGEL_APPS = {"pagerank", "shortestpath"}


def local_vertex_range(num_vertices: int, rank: int, world: int):
    lo = (num_vertices * rank) // world
    hi = (num_vertices * (rank + 1)) // world
    return lo, hi


def run_pregel_job(job: JobConfig, ctx, cp: Optional[ControlPlane] = None,
                   tus: Optional[TaskUnitScheduler] = None) -> dict:
    a = dict(num_vertices=1024, out_degree=4, num_iters=20, source=0,
             edge_weight=1.0)
    a.update(job.app_args)
    cp = cp or ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    n = int(a["num_vertices"])
    # align the vertex partition with the table partition: the engine's
    # tables use contiguous even blocks, so the local range is rank-even too
    if job.app == "pagerank":
        comp = PageRankComputation(num_iters=int(a["num_iters"]))
    elif job.app == "shortestpath":
        comp = ShortestPathComputation(source=int(a["source"]),
                                       edge_weight=float(a["edge_weight"]))
    else:
        raise KeyError(job.app)
    t0 = time.perf_counter()
    # the engine derives its shard layout from Table ownership; build tables
    # first, then the graph over the local key range
    engine = PregelEngine(job, comp, n, ctx=ctx, cp=cp, tus=tus,
                          max_supersteps=int(a.get("max_supersteps", 200)))
    lo, hi = engine.local_vertex_range()
    engine.set_graph(make_ring_plus_random_graph(
        n, int(a["out_degree"]), lo, hi, ctx.device,
        stable_seed(job.job_id, "graph")))   # rank-independent graph
    values = engine.run()
    dt = time.perf_counter() - t0
    return {
        "job_id": job.job_id,
        "rank": ctx.rank,
        "num_batches": engine.supersteps_run,     # supersteps as "batches"
        "supersteps": engine.supersteps_run,
        "num_local_vertices": int(values.shape[0]),
        "total_examples": int(values.shape[0]) * engine.supersteps_run,
        "elapsed_sec": dt,
        "data_processing_rate": (int(values.shape[0]) * engine.supersteps_run
                                 / dt if dt else 0.0),
    }
