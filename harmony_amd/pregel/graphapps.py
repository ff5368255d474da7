"""Graph apps: PageRank + single-source shortest path.

Reference: pregel/graphapps/pagerank (PagerankComputation + sum combiner)
and pregel/graphapps/shortestpath (min combiner). Both are vectorized over
the local vertex partition (see engine.Computation).
"""

from __future__ import annotations

import torch

from harmony_amd.pregel.engine import Computation, LocalGraph


class PageRankComputation(Computation):
    """val = 0.15/N + 0.85 * sum(in msgs); send val/out_deg along edges for
    `num_iters` supersteps (reference PagerankComputation)."""

    combiner = "add"
    msg_identity = 0.0
    msg_dim = 1

    def __init__(self, num_iters: int = 20):
        self.num_iters = num_iters

    def compute(self, superstep, values, incoming, has_msg, graph: LocalGraph):
        n = graph.num_vertices_global
        if superstep == 0:
            new = torch.full_like(values, 1.0 / n)
        else:
            new = 0.15 / n + 0.85 * incoming
        if superstep >= self.num_iters:
            active = torch.zeros(values.shape[0], dtype=torch.bool,
                                 device=values.device)
            return new, None, active
        counts = graph.row_ptr[1:] - graph.row_ptr[:-1]
        contrib = new.squeeze(1) / graph.out_degree.clamp_min(1)
        edge_msgs = contrib.repeat_interleave(counts).unsqueeze(1)
        active = torch.ones(values.shape[0], dtype=torch.bool,
                            device=values.device)
        return new, edge_msgs, active


class ShortestPathComputation(Computation):
    """Min-combiner relaxation: a vertex adopts the smallest incoming
    distance and, when improved, relaxes its out-edges (reference
    shortestpath app). Vertices halt immediately; messages wake them."""

    combiner = "min"
    msg_identity = float("inf")
    msg_dim = 1

    def __init__(self, source: int = 0, edge_weight: float = 1.0):
        self.source = source
        self.edge_weight = edge_weight

    def compute(self, superstep, values, incoming, has_msg, graph: LocalGraph):
        dev = values.device
        n_local = values.shape[0]
        if superstep == 0:
            new = torch.full_like(values, float("inf"))
            lo = graph.vertex_lo
            src_local = self.source - lo
            changed = torch.zeros(n_local, dtype=torch.bool, device=dev)
            if 0 <= src_local < n_local:
                new[src_local] = 0.0
                changed[src_local] = True
        else:
            cand = torch.where(has_msg.unsqueeze(1), incoming,
                               torch.full_like(incoming, float("inf")))
            new = torch.minimum(values, cand)
            changed = (new < values).any(dim=1)
        edge_msgs = None
        if bool(changed.any()):
            counts = graph.row_ptr[1:] - graph.row_ptr[:-1]
            per_edge_changed = changed.repeat_interleave(counts)
            dist_per_edge = new.squeeze(1).repeat_interleave(counts)
            msgs = torch.where(per_edge_changed,
                               dist_per_edge + self.edge_weight,
                               torch.full_like(dist_per_edge, float("inf")))
            edge_msgs = msgs.unsqueeze(1)
        active = torch.zeros(n_local, dtype=torch.bool, device=dev)
        return new, edge_msgs, active


def make_ring_plus_random_graph(num_vertices: int, out_degree: int,
                                vertex_lo: int, vertex_hi: int,
                                device, seed: int) -> LocalGraph:
    """Synthetic graph: ring edge (connectivity) + random extra edges.

    Edges are a pure function of (seed, vertex id) via the counter RNG, so
    the GLOBAL graph is identical for every world size / partitioning — a
    2-rank run computes exactly the same PageRank as a 1-rank run."""
    from harmony_amd.ops.rng import rng_u32

    n_local = vertex_hi - vertex_lo
    deg = out_degree
    vs = torch.arange(vertex_lo, vertex_hi, dtype=torch.int64)
    ring = (vs + 1) % num_vertices                       # [n_local]
    if deg > 1:
        ctr = (vs.unsqueeze(1) * (deg - 1)
               + torch.arange(deg - 1, dtype=torch.int64))  # [n_local, deg-1]
        extra = rng_u32(seed & 0xFFFFFFFF, ctr.reshape(-1)) % num_vertices
        edge_dst = torch.cat([ring.unsqueeze(1),
                              extra.view(n_local, deg - 1)], dim=1).reshape(-1)
    else:
        edge_dst = ring
    row_ptr = torch.arange(0, (n_local + 1) * deg, deg, device=device)
    out_deg = torch.full((n_local,), float(deg), device=device)
    return LocalGraph(vertex_lo=vertex_lo, row_ptr=row_ptr,
                      edge_dst=edge_dst.to(device), out_degree=out_deg,
                      num_vertices_global=num_vertices)
