"""Elasticity plan: op vocabulary + DAG executor.

Reference: et/plan/impl/ETPlan.java:36-87 (op DAG with dependency-driven
release) and plan/impl/op/* (Allocate/Deallocate/Create/Drop/Associate/
Unassociate/Subscribe/Unsubscribe/Move/Start/Stop — "the elasticity
instruction set"), compiled from the optimizer's output by
dolphin/plan/PlanCompiler.java:45.

MI355X mapping: executors are fixed GPU ranks (the pool is the node), so
Allocate/Deallocate become no-ops of the pool; the live instruction set is
  MoveOp          — migrate table blocks between GPU shards (et/migration)
  SetBatchShareOp — change each rank's share of mini-batch work (the
                    worker add/delete/switch of the reference: a rank whose
                    share drops to 0 is a pure server)
Ops execute collectively at a quiesced point; the DAG orders them (moves of
one table before share changes that depend on them, etc.).
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from harmony_amd.utils.dag import DAG


@dataclass(frozen=True)
class MoveOp:
    table_id: str
    moves: tuple          # tuple of (block_id, dst_rank) pairs

    def to_json(self):
        return {"op": "move", "table_id": self.table_id,
                "moves": list(self.moves)}


@dataclass(frozen=True)
class SetBatchShareOp:
    shares: tuple         # tuple of (rank, num_batches_per_epoch)

    def to_json(self):
        return {"op": "share", "shares": list(self.shares)}


@dataclass(frozen=True)
class StopWorkerOp:
    """Stop a rank's worker role: its batch share drops to zero — it keeps
    serving its table blocks (pure server), exactly the reference's
    worker->server switch (PlanCompiler.translateToSwitch)."""

    rank: int

    def to_json(self):
        return {"op": "stop_worker", "rank": self.rank}


@dataclass(frozen=True)
class StartWorkerOp:
    rank: int
    num_batches: int

    def to_json(self):
        return {"op": "start_worker", "rank": self.rank,
                "num_batches": self.num_batches}


@dataclass(frozen=True)
class DropTableOp:
    """Free a table's device shard on every rank (reference DropOp; the
    jobserver drops per-job tables when a job finishes)."""

    table_id: str

    def to_json(self):
        return {"op": "drop", "table_id": self.table_id}


def compile_switch(table_id: str, stop_ranks, start, moves) -> "Plan":
    """Role-switch plan compiler (reference PlanCompiler.translateToSwitch,
    dolphin/plan/PlanCompiler.java:66-90): ranks leaving the worker role
    STOP first, then their mini-batch load and/or table blocks MOVE, then
    joining ranks START — the dependency DAG encodes that order so the
    executor can never start a worker before its data exists.

    stop_ranks: ranks whose worker role ends (become pure servers)
    start:      {rank: num_batches} ranks (re)starting as workers
    moves:      {block_id: dst_rank} table blocks to migrate
    """
    p = Plan()
    stop_idx = []
    for r in sorted(stop_ranks):
        stop_idx.append(len(p.ops))
        p.ops.append(StopWorkerOp(r))
    move_idx = None
    if moves:
        move_idx = len(p.ops)
        p.ops.append(MoveOp(table_id, tuple(sorted(moves.items()))))
        for i in stop_idx:
            p.deps.append((i, move_idx))
    for r, n in sorted((start or {}).items()):
        i = len(p.ops)
        p.ops.append(StartWorkerOp(r, n))
        p.deps.append((move_idx if move_idx is not None
                       else (stop_idx[-1] if stop_idx else i), i))
        if move_idx is None and not stop_idx:
            p.deps.pop()          # nothing to depend on
    return p


def op_from_json(d: dict):
    if d["op"] == "move":
        return MoveOp(d["table_id"], tuple(tuple(m) for m in d["moves"]))
    if d["op"] == "share":
        return SetBatchShareOp(tuple(tuple(s) for s in d["shares"]))
    if d["op"] == "stop_worker":
        return StopWorkerOp(d["rank"])
    if d["op"] == "start_worker":
        return StartWorkerOp(d["rank"], d["num_batches"])
    if d["op"] == "drop":
        return DropTableOp(d["table_id"])
    raise ValueError(d)


@dataclass
class Plan:
    ops: List[object] = field(default_factory=list)
    deps: List[tuple] = field(default_factory=list)   # (before_idx, after_idx)
    estimated_benefit: float = 0.0

    def to_json(self) -> str:
        return json.dumps({"ops": [o.to_json() for o in self.ops],
                           "deps": self.deps,
                           "benefit": self.estimated_benefit})

    @staticmethod
    def from_json(s: str) -> "Plan":
        d = json.loads(s)
        return Plan(ops=[op_from_json(o) for o in d["ops"]],
                    deps=[tuple(e) for e in d["deps"]],
                    estimated_benefit=d.get("benefit", 0.0))

    def empty(self) -> bool:
        return not self.ops


class PlanExecutor:
    """Executes a plan's op DAG (reference PlanExecutorImpl.java:41 runs ready
    sets on a thread pool; here ops are collective so ready ops execute in
    deterministic order on every rank)."""

    def __init__(self, tables: Dict[str, object], rank: int, world_size: int,
                 group=None):
        # index by the table's GLOBAL id (cfg.table_id), not the app-local name
        self.tables = {t.cfg.table_id: t for t in tables.values()
                       if hasattr(t, "cfg")}
        self.rank = rank
        self.world_size = world_size
        self.group = group
        self.batch_shares: Optional[Dict[int, int]] = None

    def execute(self, plan: Plan) -> None:
        from harmony_amd.et.migration import migrate

        dag: DAG[int] = DAG()
        for i in range(len(plan.ops)):
            dag.add_vertex(i)
        for b, a in plan.deps:
            dag.add_edge(b, a)
        ready = sorted(dag.roots())
        while ready:
            i = ready.pop(0)
            op = plan.ops[i]
            if isinstance(op, MoveOp):
                table = self.tables.get(op.table_id)
                if table is not None:
                    migrate(table, dict(op.moves), self.rank,
                            self.world_size, group=self.group)
            elif isinstance(op, SetBatchShareOp):
                # MERGE (not replace): a plan can stop rank r and set the
                # remaining workers' shares — replacing here would silently
                # wipe the StopWorkerOp's share-0 applied moments earlier
                if self.batch_shares is None:
                    self.batch_shares = {}
                self.batch_shares.update(dict(op.shares))
            elif isinstance(op, StopWorkerOp):
                if self.batch_shares is None:
                    self.batch_shares = {}
                self.batch_shares[op.rank] = 0
            elif isinstance(op, StartWorkerOp):
                if self.batch_shares is None:
                    self.batch_shares = {}
                self.batch_shares[op.rank] = op.num_batches
            elif isinstance(op, DropTableOp):
                table = self.tables.pop(op.table_id, None)
                if table is not None and hasattr(table, "drop_blocks"):
                    table.drop_blocks(list(table.owned_blocks))
            ready = sorted(ready + dag.on_complete(i))
        # plans change key routing (migration moved blocks; share changes
        # resliced batches) — drop route/count caches on EVERY rank at the
        # same quiesced point, or cache hit/miss diverges across ranks and
        # desynchronizes the next push's collectives
        for t in self.tables.values():
            c = getattr(t, "comm", None)
            if c is not None and hasattr(c, "invalidate_routes"):
                c.invalidate_routes()
