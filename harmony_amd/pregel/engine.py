"""The BSP graph engine.

Reference mapping:
  * vertex table (Long -> Vertex<V,E>)      -> dense Table of vertex values;
    the adjacency (CSR over the LOCAL vertex partition) is device-resident
    per rank, like the reference's vertex edges living with the vertex.
  * two message tables, double-buffered     -> two dense Tables with a
    combiner update function ("add" for sum-combiners, "min" for min-
    combiners) — a message send IS a combining scatter on the owner
    (reference MessageManager.addMessage with MessageCombiner:92, flip:72).
  * superstep control (SuperstepResultMsg / SuperstepControlMsg,
    WorkerMsgManager.java:62-90)            -> an all-reduce of
    {num_active, msgs_sent} votes; the job halts when all vertices halted
    and no messages are in flight (PregelMaster.java:48-56).

The Computation SPI is vectorized MI355X-first: compute() transforms the
whole local vertex partition at once (values tensor + combined incoming
messages) instead of a per-vertex callable — per-vertex Java iteration
(ComputationCallable.java:36) would serialize the GPU.
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, Optional, Tuple

import torch
import torch.distributed as dist

from harmony_amd.config import JobConfig, TableConfig
from harmony_amd.et.table import Table
from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler


class Computation:
    """Vectorized vertex program (reference graph/api/Computation.java:25).

    All tensors are the LOCAL partition (rows = local vertices, aligned with
    the vertex table's local shard)."""

    # message-combiner semantics: "add" (sum) or "min"
    combiner = "add"
    # identity element written into the cleared message buffer
    msg_identity = 0.0
    msg_dim = 1

    def compute(self, superstep: int, values: torch.Tensor,
                incoming: torch.Tensor, has_msg: torch.Tensor, graph
                ) -> Tuple[torch.Tensor, Optional[torch.Tensor], torch.Tensor]:
        """Returns (new_values, per_edge_messages or None, active_mask).
        per_edge_messages: [num_local_edges, msg_dim] — message sent along
        each outgoing edge of the local CSR (dst = graph.edge_dst)."""
        raise NotImplementedError


@dataclass
class LocalGraph:
    """CSR adjacency of the local vertex partition."""

    vertex_lo: int                 # first global vertex id owned locally
    row_ptr: torch.Tensor          # [n_local+1]
    edge_dst: torch.Tensor         # [n_local_edges] global dst ids
    out_degree: torch.Tensor       # [n_local]
    num_vertices_global: int


class PregelEngine:
    def __init__(self, job: JobConfig, comp: Computation, num_vertices: int,
                 ctx, cp: ControlPlane, tus: Optional[TaskUnitScheduler] = None,
                 max_supersteps: int = 100):
        self.job = job
        self.comp = comp
        self.graph: Optional[LocalGraph] = None   # set_graph() after tables
        self.ctx = ctx
        self.cp = cp
        self.tus = tus or TaskUnitScheduler(cp, {job.job_id})
        self.max_supersteps = max_supersteps
        n = num_vertices
        mk = lambda name, init: Table(  # noqa: E731
            TableConfig(table_id=f"{job.job_id}/{name}", num_keys=n,
                        value_dim=comp.msg_dim, dtype="float32",
                        num_blocks=max(ctx.world_size, min(256, n)),
                        update_fn=comp.combiner, init_fn="zeros"),
            ctx.rank, ctx.world_size, ctx.device, comm=ctx.new_data_plane())
        self.vertex_table = Table(
            TableConfig(table_id=f"{job.job_id}/vertex", num_keys=n,
                        value_dim=comp.msg_dim, dtype="float32",
                        num_blocks=max(ctx.world_size, min(256, n)),
                        update_fn="assign", init_fn="zeros"),
            ctx.rank, ctx.world_size, ctx.device, comm=ctx.new_data_plane())
        # double-buffered message tables (reference PregelJobEntity creates
        # vertex + msg1 + msg2)
        self.msg = [mk("msg1", None), mk("msg2", None)]
        self._has = [None, None]   # has-message masks (local rows)
        self.cur = 0
        self._phase = 0
        self.supersteps_run = 0

    def local_vertex_range(self):
        """Global key range of this rank's vertex partition (the graph the
        caller builds must cover exactly this range)."""
        bs = self.vertex_table.block_size
        owned = self.vertex_table.owned_blocks
        lo = owned[0] * bs if owned else 0
        hi = min((owned[-1] + 1) * bs,
                 self.vertex_table.cfg.num_keys) if owned else 0
        return lo, hi

    def set_graph(self, graph: LocalGraph) -> None:
        self.graph = graph

    def _next_phase(self):
        self._phase += 1
        return self._phase

    def _local_slice(self, table: Table) -> torch.Tensor:
        return table.shard

    def _clear(self, table: Table, identity: float) -> None:
        table.shard.fill_(identity)

    def run(self) -> torch.Tensor:
        """Run supersteps to halt; returns final local vertex values."""
        comp, g = self.comp, self.graph
        dev = self.ctx.device
        n_local = g.row_ptr.shape[0] - 1
        values = self.vertex_table.shard[:n_local]
        ident = comp.msg_identity
        for t in self.msg:
            self._clear(t, ident)
        has_msg = torch.zeros(n_local, dtype=torch.bool, device=dev)
        active = torch.ones(n_local, dtype=torch.bool, device=dev)
        jid = self.job.job_id
        zero_keys = torch.empty(0, dtype=torch.int64, device=dev)
        zero_msgs = torch.empty((0, comp.msg_dim), dtype=torch.float32,
                                device=dev)
        lo, _hi = self.local_vertex_range()
        for step in range(self.max_supersteps):
            self.supersteps_run = step + 1
            # COMP: local vertex update over combined incoming messages
            incoming = self.msg[self.cur].shard[:n_local]
            values, edge_msgs, active = comp.compute(
                step, values, incoming, has_msg, g)
            # clear the consumed buffer for reuse (flip semantics,
            # reference MessageManager.flip:72-74)
            self._clear(self.msg[self.cur], ident)
            # SEND + SYNC fused: ONE collective per superstep. Every rank
            # enters the push even with zero messages (a per-rank skip
            # would strand peers inside the collective); the halt vote
            # (allVerticesHalt AND noOngoingMsgs, PregelMaster.java:48-56)
            # rides the push's count exchange as a piggyback sum, and the
            # data legs are skipped symmetrically when no rank sent.
            nxt = self.msg[1 - self.cur]
            keys = (g.edge_dst if edge_msgs is not None else zero_keys)
            msgs = (edge_msgs if edge_msgs is not None else zero_msgs)
            sent = int(keys.numel())
            nxt.last_touched_keys = None
            vote = torch.tensor([int(active.sum()), sent])
            with self.tus.net(jid, self._next_phase()):
                sums = nxt.update(keys, msgs, piggyback=vote)
            self.cur = 1 - self.cur
            # incremental has-message mask from the owner-apply's touched
            # keys (replaces the full-shard != identity scan)
            has_msg = torch.zeros(n_local, dtype=torch.bool, device=dev)
            touched = getattr(self.msg[self.cur], "last_touched_keys", None)
            if touched is not None and touched.numel():
                rows = self.msg[self.cur].local_rows_of(touched.to(dev))
                has_msg[rows[rows < n_local]] = True
            if sums is not None and int(sums[0]) == 0 and int(sums[1]) == 0:
                break
        self.vertex_table.shard[:n_local] = values
        return values
