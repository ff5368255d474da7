"""Optimizer SPI + implementations.

Reference: dolphin/optimizer/api/Optimizer.java:27;
HomogeneousOptimizer.java:51 (cost model: per-worker batch time t_i =
1/throughput_i + pullSize*max(1/bw_i, w/sum_server_bw); epoch cost =
numTotalMiniBatches / sum(1/t_i); greedy priority-queue block TransferStep
pairing :484-510; applied only above OptimizationBenefitThreshold);
SampleOptimizers.java:36 (canned optimizers for tests).

MI355X reading of the model: executors are collocated worker+server GPUs, so
the worker/server ratio becomes a continuous quantity — the fraction of
model blocks each GPU serves vs its share of mini-batch work. A GPU that
measures slower should serve fewer model blocks (less pull/push fan-in), so
target block counts are inversely proportional to measured per-rank batch
time. Optimizers see the live ownership maps and emit concrete MoveOps
(the reference's Optimizer->TransferStep->PlanCompiler pipeline collapsed:
moves are compiled directly against the ownership the plan will apply to).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

from harmony_amd.optimizer.plan import MoveOp, Plan


@dataclass
class RankMetrics:
    """EMA-processed per-rank window (reference MetricProcessor EMA over
    DolphinWorkerMetrics)."""

    rank: int
    batch_time_sec: float
    comp_time_sec: float
    pull_time_sec: float
    push_time_sec: float
    num_examples: int = 0


class Optimizer:
    """SPI: (metrics, table ownership maps) -> Plan (empty = no change).
    owners[table_id] = list over block ids of owning rank."""

    def optimize(self, metrics: List[RankMetrics],
                 owners: Dict[str, List[int]], world_size: int) -> Plan:
        raise NotImplementedError


def _counts(owner_list: List[int], world: int) -> List[int]:
    c = [0] * world
    for r in owner_list:
        c[r] += 1
    return c


def moves_to_targets(owner_list: List[int], target: List[int],
                     max_moves: int = 1 << 30) -> Dict[int, int]:
    """Greedy surplus->deficit pairing (reference generateTransferSteps):
    the highest block ids of each surplus rank move first (deterministic)."""
    world = len(target)
    per_rank: Dict[int, List[int]] = {r: [] for r in range(world)}
    for b, r in enumerate(owner_list):
        per_rank[r].append(b)
    surplus = {r: len(per_rank[r]) - target[r] for r in range(world)
               if len(per_rank[r]) > target[r]}
    deficit = {r: target[r] - len(per_rank[r]) for r in range(world)
               if len(per_rank[r]) < target[r]}
    moves: Dict[int, int] = {}
    for r_s in sorted(surplus):
        pool = per_rank[r_s][::-1]  # highest ids first
        for r_d in sorted(deficit):
            while (surplus[r_s] > 0 and deficit[r_d] > 0 and pool
                   and len(moves) < max_moves):
                moves[pool.pop(0)] = r_d
                surplus[r_s] -= 1
                deficit[r_d] -= 1
    return moves


class HomogeneousCostOptimizer(Optimizer):
    def __init__(self, benefit_threshold: float = 0.05,
                 max_moves_per_round: int = 256):
        self.benefit_threshold = benefit_threshold
        self.max_moves = max_moves_per_round

    def optimize(self, metrics, owners, world_size) -> Plan:
        if not metrics or world_size < 2:
            return Plan()
        comp = [max(1e-6, m.comp_time_sec) for m in metrics]
        serve = [max(0.0, m.pull_time_sec + m.push_time_sec) for m in metrics]
        plan = Plan()
        for tid, owner_list in owners.items():
            counts = _counts(owner_list, world_size)
            total = sum(counts)
            if total < world_size:
                continue
            # cost model: t_i = comp_i + kappa * blocks_i, kappa = measured
            # serve time per hosted block (the reference's pull-size/bandwidth
            # term, HomogeneousOptimizer.totalCost:461-482, with bandwidth
            # replaced by the measured per-block serve cost on xGMI)
            kappa = sum(serve) / max(1, total)
            if kappa <= 1e-9:
                continue  # no measurable serve cost -> nothing to optimize
            # equalize t_i: blocks_i = (C - comp_i)/kappa, sum = total
            C = (kappa * total + sum(comp)) / world_size
            raw = [max(0.0, (C - c) / kappa) for c in comp]
            sraw = sum(raw) or 1.0
            target = [int(round(total * r / sraw)) for r in raw]
            drift = total - sum(target)
            for i in range(abs(drift)):
                target[i % world_size] += 1 if drift > 0 else -1
            before = max(comp[i] + kappa * counts[i] for i in range(world_size))
            after = max(comp[i] + kappa * target[i] for i in range(world_size))
            gain = (before - after) / before if before > 0 else 0.0
            if gain < self.benefit_threshold:
                continue
            moves = moves_to_targets(owner_list, target, self.max_moves)
            if moves:
                plan.ops.append(MoveOp(tid, tuple(sorted(moves.items()))))
                plan.estimated_benefit = max(plan.estimated_benefit, gain)
        return plan


class SampleOptimizers:
    """Scripted optimizers (reference SampleOptimizers.java:36 — canned
    AddOneServer/DeleteOneServer/... used by the integration tests)."""

    @staticmethod
    def rotate_blocks(table_id: str, stride: int = 2):
        """Move every stride-th block to the next rank (round-robin)."""
        class _Rot(Optimizer):
            def optimize(self, metrics, owners, world_size):
                p = Plan()
                ol = owners[table_id]
                moves = tuple((b, (ol[b] + 1) % world_size)
                              for b in range(0, len(ol), stride))
                p.ops.append(MoveOp(table_id, moves))
                return p

        return _Rot()

    @staticmethod
    def concentrate(table_id: str, dst_rank: int = 0):
        """Move ALL of a table's blocks to one rank (the 'delete all servers
        but one' extreme)."""
        class _Conc(Optimizer):
            def optimize(self, metrics, owners, world_size):
                p = Plan()
                ol = owners[table_id]
                p.ops.append(MoveOp(
                    table_id, tuple((b, dst_rank) for b in range(len(ol))
                                    if ol[b] != dst_rank)))
                return p

        return _Conc()

    @staticmethod
    def batch_shares(shares):
        """Fixed per-rank mini-batch shares (exercises SetBatchShareOp
        consumption: WorkerTasklet._consume_shares -> provider.set_share)."""
        from harmony_amd.optimizer.plan import SetBatchShareOp

        class _Sh(Optimizer):
            def optimize(self, metrics, owners, world_size):
                p = Plan()
                p.ops.append(SetBatchShareOp(tuple(shares)))
                return p

        return _Sh()

    @staticmethod
    def stop_then_start(rank: int, num_batches: int = 1,
                        stop_at_call: int = 1, start_at_call: int = 3):
        """Scripted worker role switch (reference DeleteOneWorker +
        AddOneWorker): STOP `rank` at optimization round `stop_at_call`
        (its batches become EMPTY — zero examples, zero sparse wire),
        re-START it with `num_batches` share at `start_at_call`."""
        from harmony_amd.optimizer.plan import compile_switch

        class _SS(Optimizer):
            def __init__(self):
                self.calls = 0

            def optimize(self, metrics, owners, world_size):
                self.calls += 1
                if self.calls == stop_at_call:
                    return compile_switch("", [rank], {}, {})
                if self.calls == start_at_call:
                    return compile_switch("", [], {rank: num_batches}, {})
                return Plan()

        return _SS()

    @staticmethod
    def elastic_showcase(rank: int, stop_at_call: int = 1,
                         start_at_call: int = 2):
        """The BASELINE config-#5 showcase: mid-run, STOP worker `rank`
        AND live-migrate all its model blocks away (ownership-first); a
        few windows later, START it again and migrate an even share back.
        One plan per event, compiled with the reference's switch ordering
        (stop -> move -> start, PlanCompiler.translateToSwitch)."""
        from harmony_amd.optimizer.plan import compile_switch

        class _E(Optimizer):
            def __init__(self):
                self.calls = 0

            def optimize(self, metrics, owners, world_size):
                self.calls += 1
                tid = sorted(owners)[0]
                ol = owners[tid]
                if self.calls == stop_at_call:
                    others = [r for r in range(world_size) if r != rank]
                    moves = {b: others[i % len(others)] for i, b in
                             enumerate(i for i, o in enumerate(ol)
                                       if o == rank)}
                    return compile_switch(tid, [rank], {}, moves)
                if self.calls == start_at_call:
                    total = len(ol)
                    per = total // world_size
                    mine = [i for i, o in enumerate(ol) if o == rank]
                    need = per - len(mine)
                    moves = {}
                    if need > 0:
                        donors = [i for i, o in enumerate(ol) if o != rank]
                        for b in donors[:need]:
                            moves[b] = rank
                    return compile_switch(tid, [], {rank: 1}, moves)
                return Plan()

        return _E()

    @staticmethod
    def even_rebalance(table_id: str):
        class _Even(Optimizer):
            def optimize(self, metrics, owners, world_size):
                ol = owners[table_id]
                total = len(ol)
                base = total // world_size
                target = [base + (1 if r < total % world_size else 0)
                          for r in range(world_size)]
                moves = moves_to_targets(ol, target)
                p = Plan()
                if moves:
                    p.ops.append(MoveOp(table_id, tuple(sorted(moves.items()))))
                return p

        return _Even()
