"""Optimization orchestrator: collect metrics -> optimize -> publish plan ->
all ranks apply it collectively at the same batch boundary.

Reference: dolphin/optimizer/impl/ETOptimizationOrchestrator.java:36 —
background loop gathering EMA'd worker/server metrics, running the
Optimizer, compiling and executing the plan inside the
WorkerStateManager's RUN<->OPTIMIZE window.

MI355X shape: decisions are made on the job's rank 0 from per-rank metric
summaries published through the control store; the chosen plan is published
with an `apply_at` batch index (a multiple of the check period). Every rank
checks at those boundaries and executes the plan inside a NET-phase ticket —
the optimization window is the quiesced gap between two mini-batches, so no
update can straddle a migration (ownership-first correctness for free).
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional

from harmony_amd.optimizer.optimizers import Optimizer, RankMetrics
from harmony_amd.optimizer.plan import Plan, PlanExecutor
from harmony_amd.runtime.control import ControlPlane


class OptimizationOrchestrator:
    """One per (job, rank). Worker loop calls on_batch_boundary() between
    batches; rank 0 additionally decides and publishes plans."""

    def __init__(self, cp: ControlPlane, job_id: str, rank: int,
                 world_size: int, tables: Dict[str, object],
                 optimizer: Optional[Optimizer] = None,
                 check_period: int = 8, min_metrics: int = 1,
                 ema: float = 0.5, group=None, async_plane: bool = False):
        self.cp = cp
        self.job_id = job_id
        self.rank = rank
        self.world_size = world_size
        self.tables = tables
        self.optimizer = optimizer
        self.period = max(1, check_period)
        self.min_metrics = min_metrics
        self.ema_w = ema
        # async_plane (one-sided jobs): ranks are NOT lockstepped, so a
        # batch-index-aligned apply point can be passed by a fast rank
        # before rank 0 publishes (observed: one rank alone in the
        # migration barrier). Plans go through a sequential queue instead:
        # every rank applies plan #n at its NEXT boundary after seeing it —
        # same plans, same order, possibly different boundaries (safe: the
        # async plane has no collectives; migration syncs via the store).
        self.async_plane = async_plane
        self._seq_applied = 0
        self.executor = PlanExecutor(tables, rank, world_size, group=group)
        self._ema: Dict[int, RankMetrics] = {}
        self._batches = 0
        self._windows = 0
        self.applied_plans = 0

    # ---------------------------------------------------------- metric flow

    def _own_version(self) -> int:
        return sum(t.ownership.version for t in self.tables.values()
                   if hasattr(t, "ownership"))

    def report_batch(self, batch_time: float, comp: float, pull: float,
                     push: float, n_examples: int) -> None:
        self._batches += 1
        key = f"opt/{self.job_id}/m/{self.rank}/{self._batches}"
        if self._batches % self.period == 0:
            # tagged with the ownership version: metrics measured against a
            # different block layout are stale and must not steer a plan
            # (reference loadMetricValidationInfo,
            # ETOptimizationOrchestrator.java:164-166)
            self.cp.store.set(key, json.dumps(
                [batch_time, comp, pull, push, n_examples,
                 self._own_version()]))

    def _gather_metrics(self, at_batch: int) -> Optional[List[RankMetrics]]:
        out = []
        for r in range(self.world_size):
            key = f"opt/{self.job_id}/m/{r}/{at_batch}"
            if not self.cp.flag_set(key):
                return None   # skip round if any executor's window missing
            rec = json.loads(self.cp.store.get(key))
            b, c, pl, ps, n = rec[:5]
            if len(rec) > 5 and rec[5] != self._own_version():
                return None   # stale: measured under a different layout
            prev = self._ema.get(r)
            if prev is None:
                m = RankMetrics(r, b, c, pl, ps, n)
            else:  # EMA (reference MetricProcessor MetricWeightFactor)
                w = self.ema_w
                m = RankMetrics(r, w * b + (1 - w) * prev.batch_time_sec,
                                w * c + (1 - w) * prev.comp_time_sec,
                                w * pl + (1 - w) * prev.pull_time_sec,
                                w * ps + (1 - w) * prev.push_time_sec, n)
            self._ema[r] = m
            out.append(m)
        return out

    # ------------------------------------------------------------- decision

    def boundary_plan(self) -> Optional[Plan]:
        """Call between batches on EVERY rank (after report_batch). Rank 0
        runs the optimizer; returns the plan scheduled for THIS boundary if
        one exists (identical answer on every rank — the key is derived from
        the deterministic batch counter). The caller applies it with
        apply() inside a NET-phase ticket."""
        if self._batches % self.period != 0:
            return None
        at = self._batches
        if self.async_plane:
            # async ranks skew freely: "at - period" may not be reported
            # by a lagging rank yet (observed: every window skipped).
            # Decide on the LATEST window every rank has fully reported.
            if self.rank == 0 and self.optimizer is not None:
                t = self._latest_complete_window(at)
                if t is not None:
                    self._dec_win = t
                    self._decide(t, plan_key_base="")
        elif self.rank == 0 and self.optimizer is not None \
                and at > self.period:
            # decide on the PREVIOUS window's metrics: co-located ranks are
            # within one batch of each other, so the current window's
            # reports race rank 0's check (observed: perpetual skips when
            # jobs co-run); the previous window is guaranteed complete
            self._decide(at - self.period,
                         plan_key_base=f"opt/{self.job_id}/plan")
        if self.async_plane:
            # sequential queue: apply the next unseen plan, if any
            n_pub = self.cp.read(f"opt/{self.job_id}/planseq_n")
            if n_pub > self._seq_applied:
                key = f"opt/{self.job_id}/planseq/{self._seq_applied}"
                plan = Plan.from_json(self.cp.store.get(key).decode())
                self._seq_applied += 1
                if not plan.empty():
                    return plan
            return None
        my_key = f"opt/{self.job_id}/plan/{at}"
        if self.cp.flag_set(my_key):
            plan = Plan.from_json(self.cp.store.get(my_key).decode())
            if not plan.empty():
                return plan
        return None

    def next_pending_plan(self):
        """Async end-of-job drain step: the next published-but-unapplied
        plan, or None. Must be callable while a PEER is blocked inside a
        plan's migration barrier — so it never waits on the freeze key."""
        n_pub = self.cp.read(f"opt/{self.job_id}/planseq_n")
        while self._seq_applied < n_pub:
            key = f"opt/{self.job_id}/planseq/{self._seq_applied}"
            plan = Plan.from_json(self.cp.store.get(key).decode())
            self._seq_applied += 1
            if not plan.empty():
                return plan
        return None

    def finalize_mark(self) -> None:
        """Rank 0, on entering the end-of-job drain (past its last
        boundary, so no further decisions): freeze the final plan count."""
        if self.async_plane and self.rank == 0:
            self.cp.store.set(f"opt/{self.job_id}/planseq_final",
                              str(self.cp.read(f"opt/{self.job_id}/planseq_n")))

    def async_drained(self) -> bool:
        """True once the freeze is published and every frozen plan has
        been applied by THIS rank."""
        fin_key = f"opt/{self.job_id}/planseq_final"
        if not self.cp.flag_set(fin_key):
            return False
        return self._seq_applied >= int(self.cp.store.get(fin_key))

    def _latest_complete_window(self, upto: int):
        """Highest period-multiple <= upto for which EVERY rank's metric
        window exists (async plane: ranks report at their own pace)."""
        best = None
        t = getattr(self, "_dec_win", 0) + self.period
        while t <= upto:
            if all(self.cp.flag_set(f"opt/{self.job_id}/m/{r}/{t}")
                   for r in range(self.world_size)):
                best = t
                t += self.period
            else:
                break
        return best

    def apply(self, plan: Plan) -> None:
        """Collective: execute the plan (inside the caller's NET ticket)."""
        self.executor.execute(plan)
        self.applied_plans += 1

    def _decide(self, at: int, plan_key_base: str) -> None:
        self._windows += 1
        if self._windows < self.min_metrics:
            return
        metrics = self._gather_metrics(at)
        if metrics is None:
            return
        owners = {t.cfg.table_id: t.ownership.owner.tolist()
                  for t in self.tables.values() if hasattr(t, "cfg")}
        plan = self.optimizer.optimize(metrics, owners, self.world_size)
        if plan is not None and not plan.empty():
            if self.async_plane:
                # sequential queue: value first, THEN the count (readers
                # poll the count and fetch by index — same store client,
                # server applies in order)
                n = self.cp.read(f"opt/{self.job_id}/planseq_n")
                self.cp.store.set(f"opt/{self.job_id}/planseq/{n}",
                                  plan.to_json())
                self.cp.incr(f"opt/{self.job_id}/planseq_n", 1)
                return
            # applies two windows after `at` (one after the caller's
            # current boundary — see boundary_plan's at-period shift)
            self.cp.store.set(f"{plan_key_base}/{at + 2 * self.period}",
                              plan.to_json())
