"""Heterogeneous optimizer: MILP assignment of roles + batch share + blocks.

Reference: dolphin/optimizer/impl/hetero/ILPSolver.java:35 — a Gurobi MILP
over (worker, server, data-block, model-block) assignment for heterogeneous
machines, ILPPlanGenerator -> ILPPlanDescriptor. Gurobi does not exist here;
scipy.optimize.milp (HiGHS) solves the same shape of problem.

Model: rank i has measured per-example compute cost c_i (sec) and per-block
serve cost kappa (sec/block, averaged over the full pool). Two levels:

- `solve_assignment` (load balance only): continuous d_i (examples/step),
  integer m_i (model blocks), minimize bottleneck T with
      c_i * d_i + kappa * m_i <= T
      sum d_i = D_total, sum m_i = M_total.
- `solve_roles` (role selection, the reference's (w,s,d,m) dimension): adds
  binary w_i (rank works) and s_i (rank serves) with d_i <= D*w_i,
  m_i <= M*s_i, sum w_i = nw fixed. The per-block serve cost scales with the
  number of pulling workers (kappa * nw / n — the reference Homogeneous
  Optimizer's `w / sum(server_bw)` term, HomogeneousOptimizer.java:461-482),
  which is why nw is enumerated outside the MILP exactly as the reference
  enumerates candidate worker counts, each candidate solved as its own
  (now-linear) program.

The winning assignment compiles to the role-switch DAG: StopWorkerOp for
demoted ranks -> MoveOps (block re-partition) -> StartWorkerOp /
SetBatchShareOp (PlanCompiler.translateToSwitch ordering).
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np

from harmony_amd.optimizer.optimizers import (Optimizer, RankMetrics,
                                              moves_to_targets)
from harmony_amd.optimizer.plan import (MoveOp, Plan, SetBatchShareOp,
                                        StartWorkerOp, StopWorkerOp)


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


def solve_roles(comp_cost: List[float], kappa: float, total_examples: int,
                total_blocks: int, nw: int):
    """Role-selecting MILP for a FIXED active-worker count nw.
    -> (data_share, block_target, worker_flag, server_flag, T) or None."""
    from scipy.optimize import Bounds, LinearConstraint, milp

    n = len(comp_cost)
    if not 1 <= nw <= n:
        return None
    k_eff = kappa * nw / n          # serve load scales with pulling workers
    # variables: [d_i | m_i | w_i | s_i | T]
    nv = 4 * n + 1
    c = np.zeros(nv)
    c[-1] = 1.0
    A, lb, ub = [], [], []
    for i in range(n):
        r = np.zeros(nv)
        r[i] = comp_cost[i]
        r[n + i] = k_eff
        r[-1] = -1.0
        A.append(r); lb.append(-np.inf); ub.append(0.0)
        r = np.zeros(nv)                       # d_i <= D * w_i
        r[i] = 1.0
        r[2 * n + i] = -float(total_examples)
        A.append(r); lb.append(-np.inf); ub.append(0.0)
        r = np.zeros(nv)                       # m_i <= M * s_i
        r[n + i] = 1.0
        r[3 * n + i] = -float(total_blocks)
        A.append(r); lb.append(-np.inf); ub.append(0.0)
    for lo, hi, sl in ((total_examples, total_examples, slice(0, n)),
                       (total_blocks, total_blocks, slice(n, 2 * n)),
                       (nw, nw, slice(2 * n, 3 * n)),
                       (1, n, slice(3 * n, 4 * n))):
        r = np.zeros(nv)
        r[sl] = 1.0
        A.append(r); lb.append(lo); ub.append(hi)
    integrality = np.zeros(nv)
    integrality[n:4 * n] = 1
    hi = np.full(nv, np.inf)
    hi[2 * n:4 * n] = 1.0                      # w, s binary
    res = milp(c=c, constraints=LinearConstraint(np.array(A), lb, ub),
               integrality=integrality, bounds=Bounds(np.zeros(nv), hi))
    if not res.success:
        return None
    d = res.x[:n].tolist()
    m = np.rint(res.x[n:2 * n]).astype(int)
    drift = total_blocks - int(m.sum())
    serving = [i for i in range(n) if res.x[3 * n + i] > 0.5]
    for j in range(abs(drift)):
        m[serving[j % len(serving)]] += 1 if drift > 0 else -1
    w = [bool(res.x[2 * n + i] > 0.5) for i in range(n)]
    s = [bool(res.x[3 * n + i] > 0.5) for i in range(n)]
    return d, m.tolist(), w, s, float(res.x[-1])


def best_roles(comp_cost: List[float], kappa: float, total_examples: int,
               total_blocks: int, margin: float = 0.02):
    """Enumerate active-worker counts (the reference's candidate loop,
    HomogeneousOptimizer.java:127) and return the minimum-bottleneck
    solution of solve_roles. Among candidates within `margin` of the
    optimum, FEWER workers win: the LP will happily keep a 20x-slower
    machine working for a 0.4% predicted gain, which real-world batch
    overheads and variance never deliver."""
    sols = []
    for nw in range(1, len(comp_cost) + 1):
        sol = solve_roles(comp_cost, kappa, total_examples, total_blocks, nw)
        if sol is not None:
            sols.append((nw, sol))
    if not sols:
        return None
    t_min = min(s[-1] for _, s in sols)
    return next(s for _, s in sols if s[-1] <= t_min * (1.0 + margin))


class HeterogeneousOptimizer(Optimizer):
    def __init__(self, benefit_threshold: float = 0.05,
                 examples_per_step: int = 1 << 14,
                 role_select: bool = True):
        self.benefit_threshold = benefit_threshold
        self.examples_per_step = examples_per_step
        self.role_select = role_select

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
        # optimize the MEASURED per-step example total so the predicted
        # bottleneck T is commensurable with the measured batch time in
        # the benefit test (a configured constant here made T an absolute
        # fantasy the test always rejected); fall back to the configured
        # scale only when metrics carry no example counts
        total_ex = sum(m.num_examples for m in metrics) \
            or self.examples_per_step
        plan = Plan()
        if self.role_select:
            sol = best_roles(comp, kappa, total_ex, total_blocks)
            if sol is None:
                return plan
            d, m_tot, w, _s, T = sol
        else:
            sol = solve_assignment(comp, kappa, total_ex, total_blocks)
            if sol is None:
                return plan
            d, m_tot, T = sol
            w = [True] * world_size
        # current bottleneck for the benefit test
        cur = max(metrics[i].batch_time_sec for i in range(world_size))
        if cur <= 0 or (cur - T) / cur < self.benefit_threshold:
            return plan
        # role-switch DAG order (PlanCompiler.translateToSwitch): demoted
        # ranks STOP, then blocks MOVE, then shares/starts apply — a rank
        # whose share grows must already own its new blocks.
        active_now = [metrics[i].num_examples > 0 for i in range(world_size)]
        stop_idx = []
        for i in range(world_size):
            if active_now[i] and not w[i]:
                stop_idx.append(len(plan.ops))
                plan.ops.append(StopWorkerOp(i))
        # distribute each table's blocks proportionally to m_tot
        m_arr = np.array(m_tot, dtype=float)
        frac = m_arr / max(1.0, m_arr.sum())
        move_idx = []
        for tid, ol in owners.items():
            tot = len(ol)
            target = [int(round(tot * f)) for f in frac]
            drift = tot - sum(target)
            for i in range(abs(drift)):
                target[i % world_size] += 1 if drift > 0 else -1
            moves = moves_to_targets(ol, target)
            if moves:
                mi = len(plan.ops)
                plan.ops.append(MoveOp(tid, tuple(sorted(moves.items()))))
                move_idx.append(mi)
                for si in stop_idx:
                    plan.deps.append((si, mi))
        tail_idx = len(plan.ops)
        shares, starts = [], []
        for i in range(world_size):
            ni = max(1, int(round(d[i]))) if w[i] else 0
            if w[i] and not active_now[i]:
                starts.append((i, ni))
            elif w[i]:
                shares.append((i, ni))
        if shares:
            plan.ops.append(SetBatchShareOp(tuple(shares)))
        for i, ni in starts:
            plan.ops.append(StartWorkerOp(i, ni))
        for before in (move_idx or stop_idx):
            for after in range(tail_idx, len(plan.ops)):
                plan.deps.append((before, after))
        plan.estimated_benefit = (cur - T) / cur
        return plan
