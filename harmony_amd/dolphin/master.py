"""Per-job runner: builds tables/trainer/provider and runs the worker tasklet.

Reference: dolphin/core/master/DolphinMaster.java:177-193 + ETTaskRunner —
the master submits a ServerTasklet and WorkerTasklet to every executor. In
the SPMD rebuild the server role has no dedicated thread: all "server work"
is the owner-side update epilogue of push collectives (exactly as the
reference's ServerTasklet is a placeholder and real server work happens in
ET's UPDATE handling, core/server/ServerTasklet.java:28).
"""

from __future__ import annotations

from typing import Optional

import torch

from harmony_amd.config import JobConfig
from harmony_amd import mlapps
from harmony_amd.dolphin.metrics import MetricCollector
from harmony_amd.dolphin.worker import WorkerTasklet
from harmony_amd.runtime.bootstrap import ExecutorContext
from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler


def _make_optimizer(name: str):
    from harmony_amd.optimizer.optimizers import HomogeneousCostOptimizer

    if name == "homogeneous":
        return HomogeneousCostOptimizer()
    if name == "hetero_ilp":
        from harmony_amd.optimizer.hetero import HeterogeneousOptimizer

        return HeterogeneousOptimizer()
    if ":" in name:
        import importlib

        mod, cls = name.split(":", 1)
        return getattr(importlib.import_module(mod), cls)()
    raise KeyError(f"unknown optimizer '{name}'")


def _restore_tables(job: JobConfig, tables: dict) -> None:
    """Start from a model snapshot (reference ETMaster.createTable(chkpId):
    block files are rank-independent, so the restore re-partitions for
    whatever executor set this job runs on).

    `restore_chkp` forms, tried in order per table:
      "<chkp_id>"            — this job's own checkpoint
      "<src_job>/<chkp_id>"  — another job's (e.g. ModelChkpManager's
                               per-epoch snapshots); the per-table subdir is
                               keyed by the SOURCE job's table id, so the
                               lookup rewrites this job's prefix to the
                               source's.
    Raises if a table restores zero blocks — a silent no-op restore would
    masquerade as training-from-scratch."""
    from harmony_amd.dolphin.model_eval import _safe
    from harmony_amd.et.checkpoint import CheckpointManager

    cm = CheckpointManager(temp_root=job.chkp_path,
                           commit_root=job.chkp_commit_path)
    for t in tables.values():
        if not hasattr(t, "cfg"):
            continue
        short = t.cfg.table_id.split("/", 1)[-1]
        src, _, rest = job.restore_chkp.partition("/")
        candidates = [
            (job.job_id, f"{job.restore_chkp}/{_safe(t.cfg.table_id)}"),
            (job.job_id, job.restore_chkp),
        ]
        if rest:
            candidates += [
                (src, f"{rest}/{_safe(src + '/' + short)}"),
                (src, rest),
            ]
        loaded = -1
        for app_id, cid in candidates:
            if cm.exists(app_id, cid):
                loaded = cm.load_into(t, app_id, cid)
                if loaded > 0:
                    break
        if loaded <= 0:
            raise FileNotFoundError(
                f"restore_chkp {job.restore_chkp!r}: no usable checkpoint "
                f"for table {t.cfg.table_id} under {job.chkp_path} "
                f"(tried {candidates}; loaded={loaded})")


def run_job(job: JobConfig, ctx: ExecutorContext,
            cp: Optional[ControlPlane] = None,
            tus: Optional[TaskUnitScheduler] = None,
            stream: Optional[torch.cuda.Stream] = None,
            optimizer=None) -> MetricCollector:
    """Run one PS job to completion on this rank; returns this rank's metrics.

    Collective: every rank of the job's executor set must call this with the
    same JobConfig.
    """
    cp = cp or ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    tus = tus or TaskUnitScheduler(cp, {job.job_id}, multi_job=False)
    app = mlapps.get_app(job.app)
    tables, trainer, provider = app.build(job, ctx, cp)
    if job.restore_chkp:
        _restore_tables(job, tables)
    orch = None
    if optimizer is None and job.optimizer:
        optimizer = _make_optimizer(job.optimizer)
    if optimizer is not None:
        from harmony_amd.optimizer.orchestrator import OptimizationOrchestrator

        from harmony_amd.dolphin.model_accessor import OneSidedAccessor

        orch = OptimizationOrchestrator(
            cp, job.job_id, ctx.rank, ctx.world_size, tables,
            optimizer=optimizer, check_period=job.optimizer_period,
            group=getattr(ctx, "group", None),
            async_plane=isinstance(getattr(trainer, "accessor", None),
                                   OneSidedAccessor))
    tracer = None
    if job.trace_path:
        from harmony_amd.utils.tracing import Tracer

        tracer = Tracer(rank=ctx.rank, job=job.job_id,
                        out_path=f"{job.trace_path}.r{ctx.rank}.jsonl")
    dashboard = None
    if job.dashboard_url:
        from harmony_amd.dashboard import DashboardConnector

        dashboard = DashboardConnector(job.dashboard_url)
    tasklet = WorkerTasklet(job, trainer, provider, cp, tus,
                            ctx.rank, ctx.world_size, stream=stream,
                            orchestrator=orch, tracer=tracer,
                            dashboard=dashboard)
    chkp_mgr = None
    if job.model_chkp_per_epoch or job.offline_model_eval:
        # per-epoch model snapshots for offline evaluation (reference
        # ModelChkpManager, DolphinMaster.java:187-189)
        from harmony_amd.dolphin.model_eval import ModelChkpManager
        from harmony_amd.et.checkpoint import CheckpointManager

        cm = CheckpointManager(temp_root=job.chkp_path,
                               commit_root=job.chkp_commit_path)
        chkp_mgr = ModelChkpManager(cm, job.job_id, tables)
        orig_hook = trainer.on_epoch_finished

        def _hook(epoch, _orig=orig_hook, _mgr=chkp_mgr):
            _orig(epoch)
            _mgr.on_epoch_finished(epoch)

        trainer.on_epoch_finished = _hook
    metrics = tasklet.run()
    if orch is not None:
        metrics._applied_plans = orch.applied_plans
    if tracer is not None:
        tracer.flush()
    # evaluate_model may issue collectives (e.g. LDA's log-likelihood does
    # a pull_all) — under co-located jobs EVERY collective needs a global
    # NET ticket or two jobs' post-run evals can interleave on the wire.
    # All ranks reach here after the job's cleanup barrier, so a fixed
    # out-of-band phase index is requested in the same order everywhere.
    import contextlib

    _EVAL_PHASE = 1 << 30

    def _eval_net(idx):
        # async (one-sided) jobs issue kernels, not collectives: no ticket
        if tasklet.is_async:
            return contextlib.nullcontext()
        return tasklet.tus.net(job.job_id, idx)

    with _eval_net(_EVAL_PHASE):
        ev = trainer.evaluate_model()
    for k, v in (ev or {}).items():
        metrics.add_custom(k, float(v))
    if job.offline_model_eval and chkp_mgr is not None:
        from harmony_amd.dolphin.model_eval import ModelEvaluator

        evaluator = ModelEvaluator(chkp_mgr.cm, job.job_id, tables, trainer,
                                   provider)
        with _eval_net(_EVAL_PHASE + 1):
            offline = evaluator.evaluate_all(chkp_mgr.chkp_ids)
        for cid, res in offline.items():
            for k, v in (res or {}).items():
                metrics.add_custom(f"offline/{cid}/{k}", float(v))
    return metrics
