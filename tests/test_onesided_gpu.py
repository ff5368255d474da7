"""One-sided (async) xGMI data plane — GPU tests.

Two processes SHARE the single test GPU (hipIpc maps same-device memory
the same way it maps peer HBM over xGMI on an 8-GPU node), and run with
NO barriers between operations: uneven per-rank work that would deadlock
the collective data plane must complete and conserve every push.
"""

import datetime

import pytest
import torch

from tests.dist_helper import run_dist

pytestmark = pytest.mark.gpu


def _store(rank, world):
    import os

    from torch.distributed import TCPStore

    return TCPStore(os.environ["MASTER_ADDR"],
                    int(os.environ["MASTER_PORT"]), world,
                    is_master=(rank == 0),
                    timeout=datetime.timedelta(seconds=120),
                    wait_for_workers=False)


def _barrier(store, name, rank, world):
    store.add(f"b/{name}", 1)
    store.wait([f"b/{name}"])
    import time

    while int(store.get(f"b/{name}")) < world:
        time.sleep(0.005)


def _async_worker(rank, world):
    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable

    torch.cuda.set_device(0)           # both ranks share the test GPU
    store = _store(rank, world)
    cfg = TableConfig(table_id="os_t", num_keys=64, value_dim=8,
                      num_blocks=8, update_fn="add", init_fn="zeros")
    t = OneSidedTable(cfg, rank, world, torch.device("cuda"), store=store)
    _barrier(store, "alloc", rank, world)   # shards exist before export
    t.connect()
    _barrier(store, "conn", rank, world)

    # UNEVEN async work: rank 0 does 10 rounds, rank 1 does 5 — no
    # coupling whatsoever between the loops (impossible collectively)
    rounds = 10 if rank == 0 else 5
    keys = torch.arange(64, device="cuda")
    ones = torch.ones(64, 8, device="cuda")
    for _ in range(rounds):
        t.push(keys, ones)
        _ = t.pull(keys[rank::2])      # concurrent reads while pushing
    t.fence()
    _barrier(store, "done", rank, world)

    # all pushes from both ranks must have landed: every element = 15
    full = t.pull_full()
    torch.cuda.synchronize()
    ok = bool((full == 15.0).all())
    bad = (full != 15.0).nonzero()
    diag = []
    for i in range(min(8, bad.shape[0])):
        r_, c_ = int(bad[i, 0]), int(bad[i, 1])
        owner = 0 if r_ < 32 else 1     # 8 blocks of 8 keys, even split
        diag.append((r_, c_, float(full[r_, c_]), owner))
    _barrier(store, "checked", rank, world)
    t.close()
    return (ok, full.unique().tolist(), diag)


def test_async_uneven_push_pull_two_procs_conserved():
    res = run_dist(_async_worker, world=2, timeout=300)
    for ok, vals, diag in res:
        assert ok, f"values {vals}; (row, col, val, owner) {diag}"


def _async_mlr_worker(rank, world):
    """Async SSP-style training demo: dense pull_full + async grad push,
    each rank stepping at its own pace (slack unbounded)."""
    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable
    from harmony_amd.utils import stable_seed

    torch.cuda.set_device(0)
    store = _store(rank, world)
    C, F = 4, 32
    cfg = TableConfig(table_id="os_mlr", num_keys=C, value_dim=F,
                      num_blocks=C, update_fn="add", init_fn="zeros")
    t = OneSidedTable(cfg, rank, world, torch.device("cuda"), store=store)
    _barrier(store, "alloc", rank, world)
    t.connect()
    _barrier(store, "conn", rank, world)

    gw = torch.Generator().manual_seed(stable_seed("os_mlr", "truth", 0))
    W_true = torch.randn(C, F, generator=gw)   # SHARED truth across ranks
    g = torch.Generator().manual_seed(stable_seed("os_mlr", "data", rank))
    X = torch.randn(512, F, generator=g).cuda()
    y = (X @ W_true.t().cuda()).argmax(dim=1)
    keys = torch.arange(C, device="cuda")
    steps = 80 if rank == 0 else 48    # uneven pace
    for _ in range(steps):
        W = t.pull_full()              # async full pull
        logits = X @ W.t()
        p = torch.softmax(logits, dim=1)
        p[torch.arange(512, device="cuda"), y] -= 1
        grad = p.t() @ X / 512
        t.push(keys, -0.3 * grad)      # async add push
    t.fence()
    _barrier(store, "done", rank, world)
    W = t.pull_full()
    acc = float(((X @ W.t()).argmax(dim=1) == y).float().mean())
    _barrier(store, "checked", rank, world)
    t.close()
    return acc


def test_async_mlr_training_converges():
    res = run_dist(_async_mlr_worker, world=2, timeout=300)
    # both ranks trained against the shared async model: far above chance
    # (0.25); async staleness adds run-to-run variance, so the bar is
    # deliberately modest
    for acc in res:
        assert acc > 0.45, res


def _async_job_worker(rank, world):
    """Full run_job MLR in one-sided mode: the training loop issues zero
    collectives (gloo group only for setup barriers), SSP slack bounds
    worker skew."""
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda", backend="gloo"))
    job = JobConfig(job_id="os_job", app="mlr", max_num_epochs=3,
                    num_mini_batches=4, clock_slack=2,
                    app_args={"num_classes": 5, "num_features": 256,
                              "num_parts_per_class": 4, "batch_size": 512,
                              "step_size": 0.5, "one_sided": True})
    s = run_job(job, ctx).summary()
    return (s["num_batches"], s["accuracy"])


def test_async_mlr_job_one_sided():
    res = run_dist(_async_job_worker, world=2, timeout=300)
    for n, acc in res:
        assert n == 12
        assert acc > 0.5, res


def _async_lasso_worker(rank, world):
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda", backend="gloo"))
    job = JobConfig(job_id="os_lasso", app="lasso", max_num_epochs=2,
                    num_mini_batches=2,
                    app_args={"num_features": 64, "num_parts": 8,
                              "batch_size": 512, "lam": 0.02,
                              "one_sided": True})
    s = run_job(job, ctx).summary()
    return (s["num_batches"], s["mse"])


def test_async_lasso_job_one_sided():
    res = run_dist(_async_lasso_worker, world=2, timeout=300)
    for n, mse in res:
        assert n == 4
        assert mse < 2.0, res


def _async_lda_worker(rank, world):
    """Async LDA: int32 one-sided word-topic table, counts conserved."""
    import torch as T

    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda", backend="gloo"))
    job = JobConfig(job_id="os_lda", app="lda", max_num_epochs=3,
                    num_mini_batches=2,
                    app_args={"num_vocabs": 500, "num_topics": 64,
                              "tokens_per_doc": 16, "docs_per_batch": 64,
                              "one_sided": True})
    s = run_job(job, ctx).summary()
    # conservation: the global table's total count == total tokens pushed
    # by BOTH ranks (each rank: 2 blocks x 64 docs x 16 tokens)
    from harmony_amd import mlapps  # noqa: F401

    import math

    assert s["num_batches"] == 6
    assert math.isfinite(s["log_likelihood"])
    return s["num_batches"]


def test_async_lda_job_one_sided():
    res = run_dist(_async_lda_worker, world=2, timeout=300)
    assert res == [6, 6]


def _async_lda_conservation_worker(rank, world):
    """Direct conservation check on the shared int32 table after an async
    LDA run: sum of all word rows == total tokens, summary row matches."""
    import torch as T

    from harmony_amd import mlapps
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.worker import WorkerTasklet
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    ctx = init_executor(RuntimeConfig(device="cuda", backend="gloo"))
    cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    job = JobConfig(job_id="os_lda_c", app="lda", max_num_epochs=2,
                    num_mini_batches=2,
                    app_args={"num_vocabs": 400, "num_topics": 32,
                              "tokens_per_doc": 12, "docs_per_batch": 32,
                              "one_sided": True})
    app = mlapps.get_app("lda")
    tables, trainer, provider = app.build(job, ctx, cp)
    tus = TaskUnitScheduler(cp, {job.job_id})
    WorkerTasklet(job, trainer, provider, cp, tus, ctx.rank,
                  ctx.world_size).run()
    t = tables["lda_model"]
    t.fence()
    cp.barrier("os_lda_c/done", ctx.world_size)
    full = t.pull_full()
    T.cuda.synchronize()
    V = 400
    total_tokens = world * 2 * 32 * 12      # ranks x blocks x docs x tokens
    wt_total = int(full[:V].sum())
    summ_total = int(full[V].sum())
    cp.barrier("os_lda_c/checked", ctx.world_size)
    t.close()
    return (wt_total, summ_total, total_tokens)


def test_async_lda_counts_conserved():
    res = run_dist(_async_lda_conservation_worker, world=2, timeout=300)
    for wt, summ, want in res:
        assert wt == want, res
        assert summ == want, res
