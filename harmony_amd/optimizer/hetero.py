"""Heterogeneous optimizer: MILP assignment of batch share + model blocks.

Reference: dolphin/optimizer/impl/hetero/ILPSolver.java:35 — a Gurobi MILP
over (worker, server, data-block, model-block) assignment for heterogeneous
machines, ILPPlanGenerator -> ILPPlanDescriptor. Gurobi does not exist here;
scipy.optimize.milp (HiGHS) solves the same shape of problem.

Model: rank i has measured per-example compute cost c_i (sec) and per-block
serve cost kappa (sec/block, averaged). Decision variables: d_i = data share
(examples per step, continuous), m_i = model blocks (integer). Minimize the
bottleneck T with
    c_i * d_i + kappa * m_i <= T      for all i
    sum d_i = D_total, sum m_i = M_total, d_i, m_i >= 0.
The result compiles to MoveOps (block re-partition) + SetBatchShareOp.
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np

from harmony_amd.optimizer.optimizers import (Optimizer, RankMetrics,
                                              moves_to_targets)
from harmony_amd.optimizer.plan import MoveOp, Plan, SetBatchShareOp


def solve_assignment(comp_cost: List[float], kappa: float, total_examples: int,
                     total_blocks: int):
    """-> (data_share per rank, block_target per rank, bottleneck T)."""
    from scipy.optimize import LinearConstraint, milp

    n = len(comp_cost)
    # variables: [d_0..d_{n-1}, m_0..m_{n-1}, T]
    nv = 2 * n + 1
    c = np.zeros(nv)
    c[-1] = 1.0                                # minimize T
    A, lb, ub = [], [], []
    for i in range(n):
        row = np.zeros(nv)
        row[i] = comp_cost[i]
        row[n + i] = kappa
        row[-1] = -1.0
        A.append(row)
        lb.append(-np.inf)
        ub.append(0.0)                         # c_i d_i + k m_i - T <= 0
    rd = np.zeros(nv)
    rd[:n] = 1.0
    A.append(rd)
    lb.append(total_examples)
    ub.append(total_examples)
    rm = np.zeros(nv)
    rm[n:2 * n] = 1.0
    A.append(rm)
    lb.append(total_blocks)
    ub.append(total_blocks)
    integrality = np.zeros(nv)
    integrality[n:2 * n] = 1                   # m_i integer
    from scipy.optimize import Bounds

    res = milp(c=c, constraints=LinearConstraint(np.array(A), lb, ub),
               integrality=integrality,
               bounds=Bounds(np.zeros(nv), np.full(nv, np.inf)))
    if not res.success:
        return None
    d = res.x[:n]
    m = np.rint(res.x[n:2 * n]).astype(int)
    # fix rounding drift on blocks
    drift = total_blocks - int(m.sum())
    for i in range(abs(drift)):
        m[i % n] += 1 if drift > 0 else -1
    return d.tolist(), m.tolist(), float(res.x[-1])


class HeterogeneousOptimizer(Optimizer):
    def __init__(self, benefit_threshold: float = 0.05,
                 examples_per_step: int = 1 << 14):
        self.benefit_threshold = benefit_threshold
        self.examples_per_step = examples_per_step

    def optimize(self, metrics: List[RankMetrics],
                 owners: Dict[str, List[int]], world_size: int) -> Plan:
        if not metrics or world_size < 2:
            return Plan()
        # per-example compute cost + per-block serve cost from measurements
        comp = [max(1e-9, m.comp_time_sec / max(1, m.num_examples))
                for m in metrics]
        total_blocks = sum(len(ol) for ol in owners.values())
        serve = sum(m.pull_time_sec + m.push_time_sec for m in metrics)
        kappa = max(1e-9, serve / max(1, total_blocks))
        plan = Plan()
        sol = solve_assignment(comp, kappa, self.examples_per_step,
                               total_blocks)
        if sol is None:
            return plan
        d, m_tot, T = sol
        # current bottleneck for the benefit test
        cur = max(metrics[i].batch_time_sec for i in range(world_size))
        if cur <= 0 or (cur - T) / cur < self.benefit_threshold:
            return plan
        # distribute each table's blocks proportionally to m_tot
        m_arr = np.array(m_tot, dtype=float)
        frac = m_arr / max(1.0, m_arr.sum())
        for tid, ol in owners.items():
            tot = len(ol)
            target = [int(round(tot * f)) for f in frac]
            drift = tot - sum(target)
            for i in range(abs(drift)):
                target[i % world_size] += 1 if drift > 0 else -1
            moves = moves_to_targets(ol, target)
            if moves:
                plan.ops.append(MoveOp(tid, tuple(sorted(moves.items()))))
        # shares apply only AFTER the block moves: a rank whose share grows
        # must already own its new blocks (PlanCompiler's switch ordering)
        share_idx = len(plan.ops)
        plan.ops.append(SetBatchShareOp(
            tuple((i, max(1, int(round(di)))) for i, di in enumerate(d))))
        for i in range(share_idx):
            plan.deps.append((i, share_idx))
        plan.estimated_benefit = (cur - T) / cur
        return plan
