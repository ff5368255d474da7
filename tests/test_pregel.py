"""Pregel engine: PageRank vs dense power iteration; SSSP vs BFS reference."""

import torch

from tests.dist_helper import run_dist


def _dense_pagerank(edge_src, edge_dst, n, iters):
    """Independent dense reference."""
    M = torch.zeros(n, n)
    deg = torch.zeros(n)
    for s in edge_src.tolist():
        deg[s] += 1
    for s, d in zip(edge_src.tolist(), edge_dst.tolist()):
        M[d, s] += 1.0 / deg[s]
    v = torch.full((n,), 1.0 / n)
    for _ in range(iters):
        v = 0.15 / n + 0.85 * (M @ v)
    return v


def test_pagerank_single_matches_power_iteration():
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.pregel.runner import run_pregel_job
    from harmony_amd.pregel.engine import PregelEngine
    from harmony_amd.pregel.graphapps import (PageRankComputation,
                                              make_ring_plus_random_graph)
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane
    from harmony_amd.utils import stable_seed

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="pr1", app="pagerank",
                    app_args={"num_vertices": 64, "out_degree": 3,
                              "num_iters": 15})
    cp = ControlPlane(ctx.store, 0, 1)
    comp = PageRankComputation(num_iters=15)
    n = 64
    engine = PregelEngine(job, comp, n, ctx, cp)
    g = make_ring_plus_random_graph(n, 3, 0, n, ctx.device,
                                    stable_seed("pr1", "graph", 0))
    engine.set_graph(g)
    vals = engine.run().squeeze(1)
    # reference
    counts = g.row_ptr[1:] - g.row_ptr[:-1]
    src = torch.arange(n).repeat_interleave(counts)
    ref = _dense_pagerank(src, g.edge_dst, n, 15)
    assert torch.allclose(vals, ref, atol=1e-4), (vals[:5], ref[:5])
    assert abs(float(vals.sum()) - 1.0) < 1e-3


def _pagerank_2rank_worker(rank, world):
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.pregel.runner import run_pregel_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="pr2", app="pagerank",
                    app_args={"num_vertices": 128, "out_degree": 3,
                              "num_iters": 10})
    out = run_pregel_job(job, ctx)
    return out["supersteps"]


def test_pagerank_two_ranks():
    res = run_dist(_pagerank_2rank_worker, world=2, timeout=120)
    assert all(s >= 10 for s in res)


def test_shortestpath_matches_bfs():
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.pregel.engine import PregelEngine
    from harmony_amd.pregel.graphapps import (ShortestPathComputation,
                                              make_ring_plus_random_graph)
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane
    from harmony_amd.utils import stable_seed

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="sp1", app="shortestpath", app_args={})
    cp = ControlPlane(ctx.store, 0, 1)
    n = 48
    comp = ShortestPathComputation(source=0)
    engine = PregelEngine(job, comp, n, ctx, cp, max_supersteps=100)
    g = make_ring_plus_random_graph(n, 3, 0, n, ctx.device,
                                    stable_seed("sp1", "graph", 0))
    engine.set_graph(g)
    vals = engine.run().squeeze(1)
    # BFS reference (unit weights)
    import collections

    adj = collections.defaultdict(list)
    counts = (g.row_ptr[1:] - g.row_ptr[:-1]).tolist()
    src = [v for v, c in enumerate(counts) for _ in range(c)]
    for s, d in zip(src, g.edge_dst.tolist()):
        adj[s].append(d)
    dist = {0: 0}
    q = collections.deque([0])
    while q:
        u = q.popleft()
        for v in adj[u]:
            if v not in dist:
                dist[v] = dist[u] + 1
                q.append(v)
    ref = torch.tensor([float(dist.get(v, float("inf"))) for v in range(n)])
    assert torch.equal(vals, ref)


def _pr_values_worker(rank, world):
    """PageRank values must be identical for any world size (the graph is a
    pure function of (seed, vertex id))."""
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.pregel.engine import PregelEngine
    from harmony_amd.pregel.graphapps import (PageRankComputation,
                                              make_ring_plus_random_graph)
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane

    ctx = init_executor(RuntimeConfig(device="cpu"))
    n = 64
    job = JobConfig(job_id="prw", app="pagerank", app_args={})
    cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    comp = PageRankComputation(num_iters=12)
    engine = PregelEngine(job, comp, n, ctx, cp)
    lo, hi = engine.local_vertex_range()
    engine.set_graph(make_ring_plus_random_graph(n, 3, lo, hi, ctx.device,
                                                 12345))
    vals = engine.run().squeeze(1)
    return (lo, vals.tolist())


def test_pagerank_world_size_invariant():
    res2 = run_dist(_pr_values_worker, world=2, timeout=120)
    # single-process reference
    lo1, v1 = _pr_values_worker(0, 1)
    combined = []
    for lo, vals in sorted(res2):
        combined.extend(vals)
    assert len(combined) == len(v1)
    for a, b in zip(combined, v1):
        assert abs(a - b) < 1e-6
