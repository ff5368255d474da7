"""The worker tasklet: the per-rank training loop of one PS job.

Reference: dolphin/core/worker/WorkerTasklet.java:96-168 — global barrier,
per-epoch data preparation, per-batch [SYNC -> PULL -> COMP -> PUSH] with
task-unit scheduling around each phase, per-batch/epoch metric emission.

MI355X mapping:
  SYNC  -> SSP clock tick (control store), no collective
  PULL  -> RCCL collective under a global NET ticket
  COMP  -> HIP kernels on this job's stream (concurrent across jobs)
  PUSH  -> RCCL collective under a global NET ticket
"""

from __future__ import annotations

import contextlib
import time
from typing import Optional

import torch

from harmony_amd.config import JobConfig
from harmony_amd.dolphin.data_provider import TrainingDataProvider
from harmony_amd.dolphin.metrics import BatchMetrics, EpochMetrics, MetricCollector
from harmony_amd.dolphin.trainer import Trainer
from harmony_amd.runtime.control import ControlPlane, SSPClock, TaskUnitScheduler


class WorkerTasklet:
    def __init__(self, job: JobConfig, trainer: Trainer,
                 provider: TrainingDataProvider, cp: ControlPlane,
                 tus: TaskUnitScheduler, rank: int, world_size: int,
                 stream: Optional[torch.cuda.Stream] = None,
                 orchestrator=None, tracer=None, dashboard=None):
        from harmony_amd.utils.tracing import null_tracer

        self.job = job
        self.trainer = trainer
        self.provider = provider
        self.cp = cp
        self.tus = tus
        self.rank = rank
        self.world_size = world_size
        self.stream = stream
        self.orch = orchestrator   # optimizer.OptimizationOrchestrator
        self.tracer = tracer or null_tracer()
        self.dashboard = dashboard  # dashboard.DashboardConnector or None
        self.metrics = MetricCollector(job.job_id, rank)
        self._phase = 0
        self.ssp = SSPClock(cp, job.job_id, world_size, job.clock_slack)
        # Async (one-sided) jobs: pulls/pushes are xGMI kernels, not
        # collectives -> no NET tickets needed, and the SSP slack wait may
        # genuinely block (bounded-async). Collective jobs: ranks are
        # lockstepped by the collectives themselves (skew <= 1 batch), so
        # the SSP wait is vacuous AND, if allowed to block, can deadlock
        # against the global ticket order (see SSPClock.tick_and_wait).
        from harmony_amd.dolphin.model_accessor import OneSidedAccessor

        self.is_async = isinstance(getattr(trainer, "accessor", None),
                                   OneSidedAccessor)

    def _net(self, jid: str, lookahead: int = 0):
        """lookahead=1 on PULL also draws PUSH's ticket in the same store
        round-trip (both phases are guaranteed to run for every batch)."""
        if self.is_async:
            return contextlib.nullcontext()
        import os as _os

        if _os.environ.get("HARMONY_NO_LOOKAHEAD"):
            lookahead = 0
        return self.tus.net(jid, self._next_phase(), lookahead)

    def _next_phase(self) -> int:
        self._phase += 1
        return self._phase

    def _consume_shares(self) -> None:
        """SetBatchShareOp consumption: scale this rank's per-batch work to
        its share relative to the mean (reference: the plan redistributes
        training-data blocks; here batches are device-resident so a slow
        rank serves a prefix slice of each block instead)."""
        shares = getattr(self.orch.executor, "batch_shares", None)
        if not shares or self.rank not in shares:
            return   # ranks absent from the plan keep their current share
        if not hasattr(self.provider, "set_share"):
            return
        if shares[self.rank] <= 0:
            # StopWorkerOp: truly EMPTY batches — the rank keeps serving
            # its table blocks (pure server) but sheds all pull/push keys
            self.provider.set_share(0.0)
            return
        vals = [v for v in shares.values() if v > 0]
        mean = sum(vals) / len(vals)
        if mean > 0:
            self.provider.set_share(shares[self.rank] / mean)

    def run(self) -> MetricCollector:
        jid = self.job.job_id
        stream_ctx = (torch.cuda.stream(self.stream) if self.stream is not None
                      else contextlib.nullcontext())
        with stream_ctx:
            # per-epoch batch COUNTS must be identical across ranks (each
            # batch is a set of collectives on the collective plane, and
            # the SSP done-counters assume uniform counts on both planes).
            # An uneven file split can produce unequal block counts — fail
            # LOUDLY on every rank instead of hanging in a collective.
            if self.world_size > 1 and not self.is_async:
                nb = self.provider.num_batches
                mx = self.cp.agree_max(f"{jid}/nbmax", nb, self.world_size)
                mn = -self.cp.agree_max(f"{jid}/nbmin", -nb, self.world_size)
                if mx != mn:
                    raise RuntimeError(
                        f"rank {self.rank}: per-epoch batch counts differ "
                        f"across ranks ({mn}..{mx}; mine {nb}) — the input "
                        "split produced unequal block counts; rebalance the "
                        "input or lower num_worker_blocks")
            # initialize() may issue collectives (e.g. LDA's initial count
            # push) — serialize it like any NET phase.
            with self._net(jid):
                self.trainer.initialize()
            # INIT -> RUN global barrier (reference WorkerGlobalBarrier)
            self.cp.barrier(f"{jid}/run", self.world_size)
            stopped = False
            for epoch in range(self.job.max_num_epochs):
                if stopped:
                    break
                ep_t0 = time.perf_counter()
                ep_examples = 0
                for bidx, batch in enumerate(self.provider.epoch_iter(epoch)):
                    # SYNC: SSP clock (reference MiniBatchBarrier -> master)
                    if not self.ssp.tick_and_wait(self.rank,
                                                  wait=self.is_async):
                        stopped = True
                        break
                    b_t0 = time.perf_counter()
                    self.trainer.set_batch_data(batch)
                    # PULL (net-ticket wait timed separately: the measured
                    # control-plane cost of the global NET sequencer)
                    t0 = time.perf_counter()
                    with self.tracer.span("pull"):
                        cm = self._net(jid, lookahead=1)
                        cm.__enter__()
                        net_t = time.perf_counter() - t0
                        try:
                            self.trainer.pull_model()
                        finally:
                            cm.__exit__(None, None, None)
                    pull_t = time.perf_counter() - t0
                    # COMP
                    t0 = time.perf_counter()
                    with self.tracer.span("comp"):
                        self.trainer.local_compute()
                    comp_t = time.perf_counter() - t0
                    # PUSH
                    t0 = time.perf_counter()
                    with self.tracer.span("push"):
                        cm = self._net(jid)
                        t1 = time.perf_counter()
                        cm.__enter__()
                        net_t += time.perf_counter() - t1
                        try:
                            self.trainer.push_update()
                        finally:
                            cm.__exit__(None, None, None)
                    push_t = time.perf_counter() - t0
                    n = self.trainer.num_batch_examples()
                    ep_examples += n
                    bt = time.perf_counter() - b_t0
                    self.metrics.add_batch(BatchMetrics(
                        epoch_idx=epoch, batch_idx=bidx, num_examples=n,
                        batch_time_sec=bt,
                        pull_time_sec=pull_t, comp_time_sec=comp_t,
                        push_time_sec=push_t, net_wait_sec=net_t))
                    # optimization window (reference RUN<->OPTIMIZE state):
                    # a quiesced gap between batches where an elasticity
                    # plan (block migration) applies collectively
                    if self.orch is not None:
                        self.orch.report_batch(bt, comp_t, pull_t, push_t, n)
                        plan = self.orch.boundary_plan()
                        if plan is not None:
                            with self._net(jid):
                                self.orch.apply(plan)
                            self._consume_shares()
                self.trainer.on_epoch_finished(epoch)
                # Per-batch spans above are ENQUEUE-side on GPU (streams
                # pipeline); the epoch metric must reflect COMPLETED work
                # (reference dataProcessingRate is wall-clock per processed
                # items), so drain this job's stream before timing.
                if self.stream is not None:
                    self.stream.synchronize()
                ep_dt = time.perf_counter() - ep_t0
                self.metrics.add_epoch(EpochMetrics(
                    epoch_idx=epoch, num_examples=ep_examples,
                    epoch_time_sec=ep_dt))
                if self.dashboard is not None:
                    self.dashboard.send(jid, self.rank, {
                        "epoch": epoch,
                        "data_processing_rate":
                            ep_examples / ep_dt if ep_dt else 0.0,
                        "epoch_time_sec": ep_dt}, t=time.time())
            # async plan queue: drain plans published while this rank was
            # finishing (a straggler plan applied by only SOME ranks would
            # leave them alone in the migration barrier — observed flake).
            # Drain INCREMENTALLY while polling the freeze key: a peer may
            # be blocked inside plan k's migration barrier at its own
            # boundary, so plan k must be applied here BEFORE any
            # wait-for-freeze (waiting first deadlocked — observed).
            if self.orch is not None and self.orch.async_plane:
                self.orch.finalize_mark()
                it = 0
                while True:
                    plan = self.orch.next_pending_plan()
                    if plan is not None:
                        self.orch.apply(plan)
                        self._consume_shares()
                        continue
                    if self.orch.async_drained():
                        break
                    it += 1
                    if it % 200 == 0:
                        self.cp.check_failed()
                    time.sleep(0.0005)
            # RUN -> CLEANUP barrier
            self.cp.barrier(f"{jid}/cleanup", self.world_size)
            self.trainer.cleanup()
        return self.metrics
