"""Concurrent PS jobs sharing executors — the multi-tenancy core.

Two jobs (MLR + NMF) run in two tasklet threads on each of 2 ranks; the
global task-unit scheduler must order their NET phases identically on every
rank (else collectives interleave and deadlock). Mirrors the reference's
LocalTaskUnitScheduler/GlobalTaskUnitScheduler behavior
(SURVEY.md §2.1/§2.2)."""

import threading

from tests.dist_helper import run_dist


def _two_jobs_worker(rank, world):
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    jobs = [
        JobConfig(job_id="mj_mlr", app="mlr", max_num_epochs=2,
                  num_mini_batches=3,
                  app_args={"num_classes": 4, "num_features": 32,
                            "num_parts_per_class": 2, "batch_size": 64}),
        JobConfig(job_id="mj_nmf", app="nmf", max_num_epochs=2,
                  num_mini_batches=2,
                  app_args={"num_cols": 128, "rank": 8, "nnz_per_row": 4,
                            "rows_per_batch": 64}),
    ]
    tus = TaskUnitScheduler(cp, {j.job_id for j in jobs}, multi_job=True)
    results = {}
    errs = []

    def run_one(job):
        try:
            results[job.job_id] = run_job(job, ctx, cp=cp, tus=tus).summary()
        except Exception as e:  # noqa: BLE001
            import traceback

            errs.append(traceback.format_exc())

    threads = [threading.Thread(target=run_one, args=(j,)) for j in jobs]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=100)
    assert not errs, errs[0]
    return {jid: s["num_batches"] for jid, s in results.items()}


def test_two_concurrent_jobs_two_ranks():
    res = run_dist(_two_jobs_worker, world=2, timeout=180)
    for r in res:
        assert r["mj_mlr"] == 6
        assert r["mj_nmf"] == 4


def _three_jobs_worker(rank, world):
    import os
    import random
    import threading
    import time as _time

    os.environ["HARMONY_SANITIZE"] = "1"     # record + validate NET order

    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    jobs = [
        JobConfig(job_id="s3_mlr", app="mlr", max_num_epochs=3,
                  num_mini_batches=3,
                  app_args={"num_classes": 3, "num_features": 16,
                            "num_parts_per_class": 2, "batch_size": 32}),
        JobConfig(job_id="s3_nmf", app="nmf", max_num_epochs=2,
                  num_mini_batches=4,
                  app_args={"num_cols": 64, "rank": 8, "nnz_per_row": 4,
                            "rows_per_batch": 32}),
        JobConfig(job_id="s3_lda", app="lda", max_num_epochs=2,
                  num_mini_batches=2,
                  app_args={"num_vocabs": 300, "num_topics": 8,
                            "tokens_per_doc": 12, "docs_per_batch": 32}),
    ]
    tus = TaskUnitScheduler(cp, {j.job_id for j in jobs}, multi_job=True)
    errs, results = [], {}

    def run_one(job, jitter):
        try:
            # stagger starts randomly: ticket ordering must hold whatever
            # the interleaving
            _time.sleep(jitter)
            results[job.job_id] = run_job(job, ctx, cp=cp, tus=tus).summary()
        except Exception:  # noqa: BLE001
            import traceback

            errs.append(traceback.format_exc())

    rng = random.Random(rank * 7 + 1)
    ts = [threading.Thread(target=run_one, args=(j, rng.random() * 0.3))
          for j in jobs]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=180)
    assert not errs, errs[0]
    import torch.distributed as dist

    dist.barrier()
    if rank == 0:
        from harmony_amd.utils import sanitize

        violations = sanitize.validate(ctx.store, world)
        assert violations == [], violations
    return {j: s["num_batches"] for j, s in results.items()}


def test_three_concurrent_jobs_three_ranks_staggered():
    from tests.dist_helper import run_dist

    res = run_dist(_three_jobs_worker, world=3, timeout=300)
    for r in res:
        assert r == {"s3_mlr": 9, "s3_nmf": 8, "s3_lda": 4}


def _uneven_blocks_worker(rank, world):
    """Unequal per-rank batch counts (uneven input split) must fail
    LOUDLY on every rank instead of hanging in a collective."""
    from harmony_amd import mlapps
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.worker import WorkerTasklet
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    job = JobConfig(job_id="uneven", app="addvector", max_num_epochs=2,
                    num_mini_batches=4,
                    app_args={"num_keys": 32, "vector_dim": 4})
    tables, trainer, provider = mlapps.get_app("addvector").build(
        job, ctx, cp)
    if rank == 1:
        provider.blocks = provider.blocks[:-1]     # simulate uneven split
    t = WorkerTasklet(job, trainer, provider, cp,
                      TaskUnitScheduler(cp, {job.job_id}), ctx.rank,
                      ctx.world_size)
    try:
        t.run()
        return "no-error"
    except RuntimeError as e:
        return "batch counts differ" in str(e)


def test_uneven_batch_counts_fail_loudly():
    assert all(r is True for r in
               run_dist(_uneven_blocks_worker, world=2, timeout=120))
