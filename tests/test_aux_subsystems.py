"""Aux subsystems: tracing, state machine, datastorer, metric service,
dashboard, offline model eval, heterogeneous MILP optimizer."""

import time

import torch


def test_tracer_spans_nest_and_flush(tmp_path):
    from harmony_amd.utils.tracing import Tracer

    tr = Tracer(rank=0, job="j", out_path=str(tmp_path / "spans.jsonl"))
    with tr.span("pull"):
        with tr.span("route"):
            time.sleep(0.001)
    assert len(tr.spans) == 2
    inner = next(s for s in tr.spans if s.name == "route")
    assert inner.parent == "pull"
    tr.flush()
    import json

    lines = [json.loads(x) for x in open(tmp_path / "spans.jsonl")]
    assert {l["name"] for l in lines} == {"pull", "route"}
    assert all(l["dur_ms"] >= 0 for l in lines)


def test_state_machine_transitions():
    import pytest

    from harmony_amd.utils.state_machine import StateMachine

    sm = StateMachine({"INIT", "RUN", "OPTIMIZE", "CLEANUP"}, "INIT",
                      {("INIT", "RUN"), ("RUN", "OPTIMIZE"),
                       ("OPTIMIZE", "RUN"), ("RUN", "CLEANUP")})
    sm.set("RUN")
    assert sm.compare_and_set("RUN", "OPTIMIZE")
    assert not sm.compare_and_set("RUN", "CLEANUP")
    sm.set("RUN")
    with pytest.raises(ValueError):
        sm.set("INIT")
    sm.set("CLEANUP")
    assert sm.state == "CLEANUP"


def test_datastorer_roundtrip(tmp_path):
    from harmony_amd.utils.datastorer import LocalFSDataStorer

    ds = LocalFSDataStorer(root=str(tmp_path))
    ds.store("a/b.bin", b"hello")
    assert ds.load("a/b.bin") == b"hello"
    t = torch.randn(4, 4)
    ds.store("t.pt", t)
    assert torch.equal(ds.load("t.pt"), t)
    assert ds.exists("t.pt") and not ds.exists("nope")


def test_metric_service_flow():
    from harmony_amd.et.metric import (ExecutorMetricCollector, MetricManager,
                                       MetricReceiver)
    from harmony_amd.runtime.bootstrap import LocalStore
    from harmony_amd.runtime.control import ControlPlane

    store = LocalStore()
    cp = ControlPlane(store, 0, 1)
    got = []

    class Rec(MetricReceiver):
        def on_metric_msg(self, src, report):
            got.append((src, report))

    mgr = MetricManager(cp, world_size=1, receivers=[Rec()])
    col = ExecutorMetricCollector(cp, rank=0)
    col.add_custom("x", 1.5)
    col.flush()                      # collection not enabled -> dropped
    assert mgr.poll() == 0
    mgr.start_collection()
    col.set_table_stats("t1", num_blocks=8, sent_get_reqs=3, recv_bytes=100)
    col.flush()
    col.add_custom("x", 2.5)
    col.flush()
    assert mgr.poll() == 2
    assert got[0][1]["tableToStats"]["t1"]["numBlocks"] == 8
    assert got[1][1]["customMetrics"]["x"] == 2.5


def test_dashboard_post_and_query(tmp_path):
    import json
    import urllib.request

    from harmony_amd.dashboard import DashboardConnector, DashboardServer

    srv = DashboardServer(port=0, db_path=str(tmp_path / "d.db"))
    port = srv.start()
    conn = DashboardConnector(f"http://127.0.0.1:{port}")
    assert conn.send("jobA", 0, {"data_processing_rate": 123.0})
    assert conn.send("jobA", 1, {"data_processing_rate": 456.0})
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/data") as r:
        rows = json.loads(r.read())
    assert len(rows) == 2
    assert rows[0]["job_id"] == "jobA"
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/") as r:
        assert b"dashboard" in r.read()
    srv.stop()


def test_offline_model_eval(tmp_path):
    """Per-epoch snapshots + offline replay (reference ModelChkpManager +
    ModelEvaluator): accuracy over checkpoints must be non-garbage and the
    last checkpoint should be at least as good as the first."""
    from harmony_amd import mlapps
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.model_eval import ModelChkpManager, ModelEvaluator
    from harmony_amd.dolphin.worker import WorkerTasklet
    from harmony_amd.et.checkpoint import CheckpointManager
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cp = ControlPlane(ctx.store, 0, 1)
    job = JobConfig(job_id="me1", app="mlr", max_num_epochs=3,
                    num_mini_batches=2,
                    app_args={"num_classes": 4, "num_features": 32,
                              "num_parts_per_class": 2, "batch_size": 128,
                              "step_size": 0.5})
    app = mlapps.get_app("mlr")
    tables, trainer, provider = app.build(job, ctx, cp)
    cm = CheckpointManager(temp_root=str(tmp_path / "t"),
                           commit_root=str(tmp_path / "c"))
    chkp = ModelChkpManager(cm, "me1", tables)
    orig_epoch_hook = trainer.on_epoch_finished

    def hook(epoch):
        orig_epoch_hook(epoch)
        chkp.on_epoch_finished(epoch)

    trainer.on_epoch_finished = hook
    tus = TaskUnitScheduler(cp, {"me1"})
    WorkerTasklet(job, trainer, provider, cp, tus, 0, 1).run()
    assert len(chkp.chkp_ids) == 3
    trainer.evaluate_model()  # reset running stats
    ev = ModelEvaluator(cm, "me1", tables, trainer, provider)
    results = ev.evaluate_all(chkp.chkp_ids)
    accs = [results[c]["accuracy"] for c in chkp.chkp_ids]
    assert len(accs) == 3
    assert accs[-1] >= accs[0] - 0.05
    assert accs[-1] > 0.5
    # non-vacuity: a no-op snapshot load would evaluate the SAME live model
    # three times — the epoch-0 snapshot must differ from the final one
    assert accs[0] != accs[-1], accs


def test_hetero_milp_optimizer():
    from harmony_amd.optimizer.hetero import HeterogeneousOptimizer, solve_assignment
    from harmony_amd.optimizer.optimizers import RankMetrics

    d, m, T = solve_assignment([1e-5, 3e-5], kappa=1e-3,
                               total_examples=10000, total_blocks=64)
    # slower rank gets less data; block totals conserved
    assert d[0] > d[1]
    assert sum(m) == 64
    opt = HeterogeneousOptimizer(benefit_threshold=0.01, role_select=False)
    metrics = [RankMetrics(0, 0.30, 0.10, 0.02, 0.02, num_examples=8192),
               RankMetrics(1, 0.30, 0.30, 0.02, 0.02, num_examples=8192)]
    owners = {"t": [0] * 16 + [1] * 16}
    plan = opt.optimize(metrics, owners, 2)
    assert not plan.empty()
    kinds = {type(op).__name__ for op in plan.ops}
    assert "SetBatchShareOp" in kinds


def test_hetero_milp_role_selection():
    """The (w,s) dimension of the reference's ILPSolver: a much slower
    machine is demoted to pure server (worker stopped, model blocks moved
    onto it) with the stop->move->share DAG order."""
    from harmony_amd.optimizer.hetero import HeterogeneousOptimizer, best_roles
    from harmony_amd.optimizer.optimizers import RankMetrics
    from harmony_amd.optimizer.plan import (MoveOp, SetBatchShareOp,
                                            StopWorkerOp)

    # direct solver: equal machines -> everyone works
    d, m, w, s, T = best_roles([1e-6] * 4, kappa=1e-5,
                               total_examples=40000, total_blocks=64)
    assert all(w) and sum(m) == 64 and abs(sum(d) - 40000) < 1
    opt = HeterogeneousOptimizer(benefit_threshold=0.01)
    mk = lambda r, comp: RankMetrics(r, comp + 0.01, comp, 0.005, 0.005,
                                     num_examples=10000)  # noqa: E731
    metrics = [mk(0, 0.04), mk(1, 0.04), mk(2, 0.04), mk(3, 0.79)]
    owners = {"t": [0] * 16 + [1] * 16 + [2] * 16 + [3] * 16}
    plan = opt.optimize(metrics, owners, 4)
    stops = [op.rank for op in plan.ops if isinstance(op, StopWorkerOp)]
    assert stops == [3]
    share = next(op for op in plan.ops if isinstance(op, SetBatchShareOp))
    assert 3 not in {r for r, _ in share.shares}
    mv = next(op for op in plan.ops if isinstance(op, MoveOp))
    # demoted rank gains blocks (pure server), and order is stop->move->share
    assert any(dst == 3 for _, dst in mv.moves)
    si = plan.ops.index(next(o for o in plan.ops
                             if isinstance(o, StopWorkerOp)))
    mi, shi = plan.ops.index(mv), plan.ops.index(share)
    assert (si, mi) in plan.deps and (mi, shi) in plan.deps


def test_offline_eval_via_job_flags(tmp_path):
    """-model_chkp_per_epoch/-offline_model_eval wired through run_job."""
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    job = JobConfig(job_id="oev", app="mlr", max_num_epochs=2,
                    num_mini_batches=2, model_chkp_per_epoch=True,
                    offline_model_eval=True, chkp_path=str(tmp_path),
                    app_args={"num_classes": 3, "num_features": 16,
                              "num_parts_per_class": 2, "batch_size": 64,
                              "step_size": 0.5})
    ctx = init_executor(RuntimeConfig(device="cpu"))
    s = run_job(job, ctx).summary()
    assert "offline/epoch0/accuracy" in s
    assert "offline/epoch1/accuracy" in s
    # non-vacuity: distinct snapshots -> distinct replayed accuracies
    assert s["offline/epoch0/accuracy"] != s["offline/epoch1/accuracy"]


def test_sanitizer_validate_unit():
    # in-memory store stub with append/get
    from harmony_amd.utils import sanitize

    class _S:
        def __init__(self):
            self.d = {}

        def append(self, k, v):
            self.d[k] = self.d.get(k, "") + v

        def get(self, k):
            return self.d[k].encode()

    s = _S()
    # clean: both ranks agree on shared-job tickets, monotone per rank
    sanitize.record(s, 0, "jA", 1, 1)
    sanitize.record(s, 0, "jB", 1, 2)
    sanitize.record(s, 1, "jA", 1, 1)
    sanitize.record(s, 1, "jB", 1, 2)
    assert sanitize.validate(s, 2) == []
    # divergence: rank 1 drew a different ticket for jB@2 than rank 0
    sanitize.record(s, 0, "jB", 2, 3)
    sanitize.record(s, 1, "jB", 2, 4)
    errs = sanitize.validate(s, 2)
    assert errs and "different orders" in errs[0]


def _san_dist_worker(rank, world):
    import os
    import threading

    os.environ["HARMONY_SANITIZE"] = "1"
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler
    from harmony_amd.utils import sanitize

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
    jobs = [JobConfig(job_id="sa_mlr", app="mlr", max_num_epochs=2,
                      num_mini_batches=2,
                      app_args={"num_classes": 3, "num_features": 16,
                                "num_parts_per_class": 2, "batch_size": 32}),
            JobConfig(job_id="sa_nmf", app="nmf", max_num_epochs=2,
                      num_mini_batches=2,
                      app_args={"num_cols": 64, "rank": 8, "nnz_per_row": 4,
                                "rows_per_batch": 32})]
    tus = TaskUnitScheduler(cp, {j.job_id for j in jobs}, multi_job=True)
    errs = []

    def run_one(j):
        try:
            run_job(j, ctx, cp=cp, tus=tus)
        except Exception:  # noqa: BLE001
            import traceback

            errs.append(traceback.format_exc())

    ts = [threading.Thread(target=run_one, args=(j,)) for j in jobs]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=100)
    assert not errs, errs[0]
    import torch.distributed as dist

    dist.barrier()
    return sanitize.validate(ctx.store, world) if rank == 0 else []


def test_sanitizer_multi_job_two_ranks_clean():
    # HARMONY_SANITIZE=1 on a real 2-rank 2-job run: the recorded NET
    # order must satisfy the deadlock-freedom invariants
    from tests.dist_helper import run_dist

    res = run_dist(_san_dist_worker, world=2, timeout=180)
    assert res[0] == []


def test_dashboard_jobs_index_and_filter(tmp_path):
    import json
    import urllib.request

    from harmony_amd.dashboard.server import (DashboardConnector,
                                              DashboardServer)

    srv = DashboardServer(db_path=str(tmp_path / "d.db"))
    port = srv.start()
    try:
        c = DashboardConnector(f"http://127.0.0.1:{port}")
        assert c.send("jobA", 0, {"data_processing_rate": 10.0}, t=1.0)
        assert c.send("jobA", 0, {"data_processing_rate": 20.0}, t=2.0)
        assert c.send("jobB", 0, {"data_processing_rate": 5.0}, t=3.0)
        jobs = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{port}/jobs", timeout=5).read())
        assert [j["job_id"] for j in jobs] == ["jobA", "jobB"]
        assert jobs[0]["reports"] == 2 and jobs[0]["t0"] == 1.0
        only_b = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{port}/data?job=jobB", timeout=5).read())
        assert len(only_b) == 1 and only_b[0]["job_id"] == "jobB"
    finally:
        srv.stop()


def test_one_sided_requires_gpu_clear_error():
    import pytest as _pt
    import torch

    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable

    if torch.cuda.is_available():
        _pt.skip("CPU-only check")
    cfg = TableConfig(table_id="x", num_keys=8, value_dim=2, num_blocks=2,
                      update_fn="add", init_fn="zeros")
    with _pt.raises(RuntimeError, match="HIP extension"):
        OneSidedTable(cfg, 0, 1, torch.device("cpu"))


def test_ssp_clock_bounds_skew_and_stops():
    # reference MiniBatchController: a fast worker blocks when more than
    # `slack` batches ahead of the slowest; stop_at halts deterministically
    import threading
    import time

    from harmony_amd.runtime.bootstrap import LocalStore
    from harmony_amd.runtime.control import ControlPlane, SSPClock

    store = LocalStore()
    cp = ControlPlane(store, 0, 1)
    clock = SSPClock(cp, "ssp_t", num_workers=2, slack=1)
    progress = {0: 0, 1: 0}
    max_skew = [0]
    done = threading.Event()

    def fast():
        for _ in range(20):
            assert clock.tick_and_wait(0)
            progress[0] += 1
            max_skew[0] = max(max_skew[0], progress[0] - progress[1])
        done.set()

    def slow():
        for _ in range(20):
            time.sleep(0.002)
            assert clock.tick_and_wait(1)
            progress[1] += 1

    ts = [threading.Thread(target=fast), threading.Thread(target=slow)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=30)
    assert done.is_set()
    # skew observed between a tick and the peer's next tick can be at most
    # slack+1 (the fast worker passes the gate, increments, THEN we sample)
    assert max_skew[0] <= 2, max_skew[0]

    # deterministic stop: both workers stop after exactly 3 more batches
    clock2 = SSPClock(cp, "ssp_s", num_workers=2, slack=-1)
    clock2.request_stop_at(3)
    for r in (0, 1):
        ticks = 0
        while clock2.tick_and_wait(r):
            ticks += 1
        assert ticks == 3, ticks


def test_task_unit_scheduler_orders_interleaved_jobs():
    # direct unit test of the ticket protocol (integration tests cover the
    # full stack): two "jobs" on two threads of one rank must execute NET
    # phases in global-ticket order even when requested out of order
    import threading
    import time

    from harmony_amd.runtime.bootstrap import LocalStore
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    store = LocalStore()
    cp = ControlPlane(store, 0, 1)
    tus = TaskUnitScheduler(cp, {"jA", "jB"}, multi_job=True)
    log = []
    lock = threading.Lock()

    def run_job(jid, phases, delay):
        for p in range(phases):
            time.sleep(delay)
            with tus.net(jid, p):
                with lock:
                    log.append((jid, p))
                time.sleep(0.001)

    ts = [threading.Thread(target=run_job, args=("jA", 6, 0.0)),
          threading.Thread(target=run_job, args=("jB", 6, 0.003))]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=30)
    assert len(log) == 12
    # reconstruct each phase's global ticket and check the log is sorted
    tickets = [int(store.get(f"tu/seq_of/{j}/{p}/v")) for j, p in log]
    assert tickets == sorted(tickets), list(zip(log, tickets))
    # per-job phases in order
    for jid in ("jA", "jB"):
        ph = [p for j, p in log if j == jid]
        assert ph == sorted(ph)


def test_heartbeat_failfast_logic():
    # failure detection (reference fail-fast handlers): a stale heartbeat
    # flips the failed+shutdown flags; fresh heartbeats do not
    import time

    from harmony_amd.jobserver.server import JobServerDriver
    from harmony_amd.runtime.bootstrap import LocalStore
    from harmony_amd.runtime.control import ControlPlane

    class _Ctx:
        rank = 0
        world_size = 2
        is_master = True
        store = LocalStore()

    drv = JobServerDriver.__new__(JobServerDriver)
    drv.ctx = _Ctx()
    drv.hb_period = 0.2
    drv.cp = ControlPlane(_Ctx.store, 0, 2)
    # rank 1's heartbeat is ancient -> the loop must fail fast and return
    _Ctx.store.set("js/hb/1", str(time.time() - 120))
    drv._heartbeat_loop()
    assert drv.cp.flag_set("js/failed")
    assert drv.cp.flag_set("js/shutdown")


def test_job_logger_prefix(capsys):
    import logging

    from harmony_amd.utils.joblog import job_logger

    lg = job_logger("jx", 2)
    assert isinstance(lg, logging.Logger)
    lg.info("hello %d", 7)
    err = capsys.readouterr().err
    assert "[jx r2] INFO hello 7" in err
    # same logger instance on repeat (no duplicate handlers)
    assert job_logger("jx", 2) is lg and len(lg.handlers) == 1


def test_failfast_unwinds_control_waits():
    """js/failed raises JobCancelled from barrier/SSP waits (VERDICT r01
    weak #7: a wedged collective must not stall shutdown for minutes)."""
    import pytest
    import torch.distributed as dist

    from harmony_amd.runtime.control import (ControlPlane, JobCancelled,
                                             SSPClock)

    store = dist.HashStore()
    cp = ControlPlane(store, 0, 2)
    cp.set_flag("js/failed")
    with pytest.raises(JobCancelled):
        cp.barrier("never", 2)
    clock = SSPClock(cp, "j", num_workers=2, slack=0)
    # rank 1 never ticks -> rank 0 is 1 ahead and would spin forever
    with pytest.raises(JobCancelled):
        clock.tick_and_wait(0)


def test_checkpoint_commit_on_close(tmp_path):
    """Temp checkpoints registered during a run move to the commit root on
    executor close (reference ChkpManagerSlave.commitAllLocalChkps:226)."""
    import torch

    from harmony_amd.config import TableConfig
    from harmony_amd.et import checkpoint as ckp
    from harmony_amd.et.table import Table

    ckp._pending_commits.clear()   # isolate from earlier tests' jobs
    cm = ckp.CheckpointManager(temp_root=str(tmp_path / "t"),
                               commit_root=str(tmp_path / "c"))
    cfg = TableConfig(table_id="cc/model", num_keys=32, value_dim=4,
                      num_blocks=4, update_fn="add", init_fn="zeros")
    t = Table(cfg, 0, 1, torch.device("cpu"))
    t.shard += 3.0
    cm.checkpoint(t, "appA", "epoch0/model")
    ckp.register_pending_commit(cm, "appA", "epoch0/model")
    assert ckp.commit_all_pending() == 1
    assert not (tmp_path / "t" / "appA" / "epoch0" / "model").exists()
    assert (tmp_path / "c" / "appA" / "epoch0" / "model").exists()
    # restore finds the committed copy (exists() checks both roots)
    t2 = Table(cfg, 0, 1, torch.device("cpu"))
    assert cm.load_into(t2, "appA", "epoch0/model") == 4
    assert torch.equal(t2.shard, t.shard)
    # registry drained: second call is a no-op
    assert ckp.commit_all_pending() == 0


def _agree_max_worker(rank, world):
    import torch.distributed as dist

    from harmony_amd.runtime.control import ControlPlane

    dist.init_process_group("gloo")
    store = dist.distributed_c10d._get_default_store()
    cp = ControlPlane(store, rank, world)
    a = cp.agree_max("t1", 10 + rank)          # max = 10 + world - 1
    b = cp.agree_max("t1", 100 - rank)         # epoch 2, same name: max=100
    return (a, b)


def test_agree_max_multiproc():
    from tests.dist_helper import run_dist

    outs = run_dist(_agree_max_worker, world=3)
    assert all(o == (12, 100) for o in outs)


def test_cached_one_sided_accessor_background_refresh():
    """Background timer refresh (reference CachedModelAccessor refresh
    thread) against a fake async table — CPU-testable thread logic."""
    import time as _time

    from harmony_amd.dolphin.model_accessor import CachedOneSidedAccessor

    class FakeCfg:
        update_fn = "add"
        update_args = {}

    class FakeTable:
        cfg = FakeCfg()

        def __init__(self):
            self.version = torch.zeros(4, 2)

        def pull_full(self):
            return self.version.clone()

        def push(self, keys, deltas):
            self.version[keys] += deltas

    t = FakeTable()
    acc = CachedOneSidedAccessor(t, refresh_sec=0.05)
    try:
        v0 = acc.pull_all()
        assert float(v0.sum()) == 0.0
        # a REMOTE writer changes the table; the refresh thread picks it up
        t.version += 1.0
        deadline = _time.monotonic() + 5
        while float(acc.pull_all().sum()) == 0.0:
            assert _time.monotonic() < deadline, "refresh thread never ran"
            _time.sleep(0.02)
        assert acc.refreshes >= 1
        # write-through push visible immediately (no refresh wait)
        acc.push(torch.tensor([0]), torch.ones(1, 2) * 5)
        assert float(acc.pull_all()[0, 0]) >= 6.0
    finally:
        acc.close()


def test_onesided_check_failed_unit():
    """OneSidedTable._check_failed raises JobCancelled when the jobserver
    failed-fast flag is set or the store is gone (ring backpressure and
    migration barriers poll it so a dead peer can't wedge a writer)."""
    import pytest

    from harmony_amd.et.onesided import OneSidedTable
    from harmony_amd.runtime.control import JobCancelled

    class FakeStore:
        def __init__(self):
            self.flag = False
            self.dead = False

        def check(self, keys):
            if self.dead:
                raise RuntimeError("connection reset")
            return self.flag

    t = OneSidedTable.__new__(OneSidedTable)   # method under test only
    t.store = FakeStore()
    t._failed_key = "js/failed"
    t._check_failed()                           # healthy: no raise
    t.store.flag = True
    with pytest.raises(JobCancelled):
        t._check_failed()
    t.store.flag = False
    t.store.dead = True                         # store gone == failed
    with pytest.raises(JobCancelled):
        t._check_failed()
