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


def _finish(store, rank, world):
    """End-of-worker sync: rank 0 hosts the TCPStore server, so it must
    exit LAST — non-zero ranks signal and leave; rank 0 polls its own
    server until everyone signalled (exiting earlier kills the server
    under a peer still polling the final barrier — observed flake)."""
    import time

    if rank == 0:
        while int(store.add("bye", 0)) < world - 1:
            time.sleep(0.005)
    else:
        store.add("bye", 1)


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
    _finish(store, rank, world)
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
    _finish(store, rank, world)
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
    _finish(ctx.store, ctx.rank, ctx.world_size)
    return (wt_total, summ_total, total_tokens)


def test_async_lda_counts_conserved():
    res = run_dist(_async_lda_conservation_worker, world=2, timeout=300)
    for wt, summ, want in res:
        assert wt == want, res
        assert summ == want, res


def _ring_worker(rank, world):
    """One-sided v2 apply-queue rings: exact totals under concurrent
    remote pushes, single-writer FIFO order through clamp nonlinearity,
    forced wraparound (cap=32 << items pushed) and backpressure."""
    import time

    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable

    torch.cuda.set_device(0)
    store = _store(rank, world)
    step, maxv = 0.1, 1e6
    cfg = TableConfig(table_id="os_ring", num_keys=32, value_dim=8,
                      num_blocks=8, update_fn="nmf_sgd",
                      update_args={"step_size": step, "max_val": maxv},
                      init_fn="zeros")
    t = OneSidedTable(cfg, rank, world, torch.device("cuda"), store=store,
                      ring_capacity=32)
    assert t._ring_mode
    _barrier(store, "alloc", rank, world)
    t.connect()
    _barrier(store, "conn", rank, world)
    other = 1 - rank
    # keys owned by the OTHER rank (pure ring path)
    all_keys = torch.arange(32, device="cuda")
    owners = t._owner_of(t.part.block_of(all_keys))
    rkeys = all_keys[owners == other]
    n = rkeys.numel()
    assert n == 16
    # phase 1: uneven rounds of remote pushes; drain own ring as we go.
    # delta -d => value += step*d per push (linear region, no clamp)
    rounds = 40 if rank == 0 else 17
    d = torch.full((n, 8), -(0.01 * (rank + 1)), device="cuda")
    for _ in range(rounds):
        t.push(rkeys, d)
        t.drain()
    t.fence()          # my enqueued remote pushes must be COMPLETE before
    store.add("pushed_done", 1)   # the peer's final-drain loop can trust 0
    while int(store.add("pushed_done", 0)) < world:
        t.drain()
        time.sleep(0.002)
    # drain until empty twice in a row (peer's last items may be in flight)
    empties = 0
    while empties < 3:
        empties = empties + 1 if t.drain() == 0 else 0
        time.sleep(0.002)
    t.fence()
    _barrier(store, "ph1", rank, world)
    # my rows got `other_rounds` pushes of delta -(0.01*(other+1))
    other_rounds = 40 if other == 0 else 17
    mine = all_keys[owners == rank]
    got = t.pull(mine)
    exp = torch.full_like(got, step * 0.01 * (other + 1) * other_rounds)
    assert torch.allclose(got, exp, atol=1e-4), (rank, got[0, 0].item(),
                                                 exp[0, 0].item())
    _barrier(store, "ph1ok", rank, world)

    # phase 2: single-writer FIFO order through the clamp: rank 1 pushes
    # an order-sensitive sequence to ONE rank-0 key; rank 0 drains then
    # both check against the in-order oracle.
    key0 = all_keys[owners == 0][:1]
    seq = [-30.0, +25.0, -1.0]   # clamp at 0 hits iff applied in order
    if rank == 1:
        for v in seq:
            t.push(key0, torch.full((1, 8), v, device="cuda"))
        t.fence()
    store.add("ph2_pushed", 1)
    if rank == 0:
        while int(store.add("ph2_pushed", 0)) < 1:
            time.sleep(0.002)
        while t.drain() == 0:
            time.sleep(0.002)
        empties = 0
        while empties < 3:
            empties = empties + 1 if t.drain() == 0 else 0
            time.sleep(0.002)
        v = float(t.pull(key0)[0, 0])
        # oracle: v0 = phase-1 value; then clamp(v - 0.1*d) in order
        v0 = step * 0.01 * 2 * 17   # rank1 pushed 17 rounds of -0.02
        for dd in seq:
            v0 = min(max(v0 - step * dd, 0.0), maxv)
        assert abs(v - v0) < 1e-4, (v, v0)
    _barrier(store, "ph2ok", rank, world)

    # phase 3: backpressure — rank 1 pushes 4x capacity while rank 0
    # delays draining; the writer must block (not corrupt) and finish.
    if rank == 1:
        for i in range(8):
            t.push(key0.repeat(16), torch.full((16, 8), -0.5, device="cuda"))
        t.fence()
    else:
        time.sleep(0.4)
        applied = 0
        while applied < 8 * 16:
            applied += t.drain()
            time.sleep(0.01)
    _barrier(store, "ph3", rank, world)
    if rank == 0:
        v = float(t.pull(key0)[0, 0])
        # 128 pushes of -0.5 after phase 2: linear adds of +0.05 each
        assert v > 0.05 * 127, v
    _barrier(store, "done", rank, world)
    t.close()
    _finish(store, rank, world)
    return True


def test_onesided_ring_apply_queue():
    assert all(run_dist(_ring_worker, world=2, timeout=300))


def _async_nmf_worker(rank, world):
    """Full run_job NMF in one-sided mode: nmf_sgd pushes go through the
    v2 apply-queue rings (non-add update fn), training converges."""
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda", backend="gloo"))
    job = JobConfig(job_id="os_nmf", app="nmf", max_num_epochs=4,
                    num_mini_batches=2, clock_slack=2,
                    app_args={"num_cols": 256, "rank": 16, "nnz_per_row": 8,
                              "rows_per_batch": 256, "step_size": 0.05,
                              "one_sided": True})
    s = run_job(job, ctx).summary()
    return (s["num_batches"], s.get("sq_err", None))


def test_async_nmf_job_one_sided_rings():
    res = run_dist(_async_nmf_worker, world=2, timeout=300)
    for n, sq in res:
        assert n == 8
        assert sq is not None


def _ring_big_worker(rank, world):
    """Pushes larger than ring capacity chunk + mutually backpressure
    without deadlock (both ranks push 4x cap at each other)."""
    import time

    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable

    torch.cuda.set_device(0)
    store = _store(rank, world)
    cfg = TableConfig(table_id="os_big", num_keys=64, value_dim=4,
                      num_blocks=8, update_fn="nmf_sgd",
                      update_args={"step_size": 1.0, "max_val": 1e9},
                      init_fn="zeros")
    t = OneSidedTable(cfg, rank, world, torch.device("cuda"), store=store,
                      ring_capacity=64)
    _barrier(store, "alloc", rank, world)
    t.connect()
    _barrier(store, "conn", rank, world)
    other = 1 - rank
    all_keys = torch.arange(64, device="cuda")
    owners = t._owner_of(t.part.block_of(all_keys))
    rkeys = all_keys[owners == other]
    # one oversized push: 4x capacity (8 repeats of 32 keys = 256 items)
    big_k = rkeys.repeat(8)
    big_d = torch.full((big_k.shape[0], 4), -1.0, device="cuda")
    t.push(big_k, big_d)          # chunks at cap=64, mutual backpressure
    t.fence()
    store.add("pushed_done", 1)
    while int(store.add("pushed_done", 0)) < world:
        t.drain()
        time.sleep(0.002)
    empties = 0
    while empties < 3:
        empties = empties + 1 if t.drain() == 0 else 0
        time.sleep(0.002)
    t.fence()
    _barrier(store, "ok", rank, world)
    mine = all_keys[owners == rank]
    got = t.pull(mine)
    # 8 pushes of -1 with step 1 => value 8.0 everywhere
    assert torch.allclose(got, torch.full_like(got, 8.0)), got[:2]
    _barrier(store, "done", rank, world)
    t.close()
    _finish(store, rank, world)
    return True


def test_onesided_ring_oversized_push_no_deadlock():
    assert all(run_dist(_ring_big_worker, world=2, timeout=300))



def _migrate_worker(rank, world):
    """Live migration of a one-sided table (v3 migrate_blocks): values
    survive the remap exactly, ownership flips on every rank, and pushes
    keep conserving afterwards — for both the atomic (add) and the
    apply-queue-ring (nmf_sgd) planes."""
    import time

    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable

    torch.cuda.set_device(0)
    store = _store(rank, world)
    dev = torch.device("cuda")

    # ---- add-mode table
    cfg = TableConfig(table_id="os_miga", num_keys=64, value_dim=4,
                      num_blocks=8, update_fn="add", init_fn="zeros")
    t = OneSidedTable(cfg, rank, world, dev, store=store)
    _barrier(store, "a_alloc", rank, world)
    t.connect()
    _barrier(store, "a_conn", rank, world)
    keys = torch.arange(64, device=dev)
    ones = torch.ones(64, 4, device=dev)
    t.push(keys, ones)
    t.fence()
    _barrier(store, "a_p1", rank, world)
    moves = {0: 1, 1: 1, 7: 0}        # 0,1 leave rank 0; 7 leaves rank 1
    t.migrate_blocks(moves)
    assert t.ownership.owner_of_int(0) == 1
    assert t.ownership.owner_of_int(7) == 0
    # both ranks' pre-migration pushes survived the remap exactly
    full = t.pull_full()
    torch.cuda.synchronize()
    assert bool((full == 2.0).all()), full.unique().tolist()
    # post-migration pushes route by the NEW ownership and conserve
    t.push(keys, ones)
    t.fence()
    _barrier(store, "a_p2", rank, world)
    full = t.pull_full()
    torch.cuda.synchronize()
    assert bool((full == 4.0).all()), full.unique().tolist()
    # moved blocks live in the adopter's local shard now
    if rank == 1:
        assert 0 in t.owned_blocks and 1 in t.owned_blocks
    else:
        assert 7 in t.owned_blocks and 0 not in t.owned_blocks
    _barrier(store, "a_ok", rank, world)
    t.close()

    # ---- ring-mode table (arbitrary update fn through apply queues)
    cfgr = TableConfig(table_id="os_migr", num_keys=32, value_dim=8,
                       num_blocks=8, update_fn="nmf_sgd", init_fn="zeros",
                       update_args={"step_size": 0.1, "max_val": 100.0})
    tr = OneSidedTable(cfgr, rank, world, dev, store=store)
    _barrier(store, "r_alloc", rank, world)
    tr.connect()
    _barrier(store, "r_conn", rank, world)
    rkeys = torch.arange(32, device=dev)
    d = torch.full((32, 8), -1.0, device=dev)   # value += 0.1 per push
    tr.push(rkeys, d)                 # remote halves queue in rings
    tr.fence()
    store.add("r_pushed", 1)
    while int(store.add("r_pushed", 0)) < world:
        tr.drain()
        time.sleep(0.002)
    # migrate_blocks drains the queues itself at the quiesce point
    tr.migrate_blocks({2: 1 - (2 % world), 5: 0})
    full = tr.pull_full()
    torch.cuda.synchronize()
    assert torch.allclose(full, torch.full_like(full, 0.1 * world),
                          atol=1e-5), full.unique().tolist()
    # ring counters restarted: a fresh round of pushes still applies
    tr.push(rkeys, d)
    tr.fence()
    store.add("r_pushed2", 1)
    while int(store.add("r_pushed2", 0)) < world:
        tr.drain()
        time.sleep(0.002)
    empties = 0
    while empties < 3:
        empties = empties + 1 if tr.drain() == 0 else 0
        time.sleep(0.002)
    _barrier(store, "r_done", rank, world)
    full = tr.pull_full()
    torch.cuda.synchronize()
    assert torch.allclose(full, torch.full_like(full, 0.2 * world),
                          atol=1e-5), full.unique().tolist()
    _barrier(store, "r_ok", rank, world)
    tr.close()
    _finish(store, rank, world)
    return True


def test_onesided_live_migration():
    assert all(run_dist(_migrate_worker, world=2, timeout=300))


def _async_migrate_job_worker(rank, world):
    """Full-stack v3: an orchestrator plan (MoveOp) migrates a ONE-SIDED
    table live inside a real async job — PlanExecutor -> et/migration
    delegation -> OneSidedTable.migrate_blocks (quiesce, one-sided block
    reads, generation remap) — and training converges through the remap."""
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.optimizer.optimizers import SampleOptimizers
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda", backend="gloo"))
    job = JobConfig(job_id="os_mlr_mig", app="mlr", max_num_epochs=6,
                    num_mini_batches=2, optimizer_period=2,
                    app_args={"num_classes": 4, "num_features": 32,
                              "num_parts_per_class": 4, "batch_size": 512,
                              "step_size": 0.3, "one_sided": True})
    opt = SampleOptimizers.rotate_blocks("os_mlr_mig/mlr_model", stride=2)
    m = run_job(job, ctx, optimizer=opt)
    s = m.summary()
    return (s["num_batches"], s.get("accuracy"),
            getattr(m, "_applied_plans", 0))


def test_async_job_live_migration():
    res = run_dist(_async_migrate_job_worker, world=2, timeout=300)
    for n, acc, plans in res:
        assert n == 12, res
        assert acc is not None and acc > 0.5, res
        assert plans >= 1, res
