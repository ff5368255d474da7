"""GPU end-to-end runs for GBT, Lasso, and Pregel apps."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_lasso_gpu():
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda"))
    job = JobConfig(job_id="g_lasso", app="lasso", max_num_epochs=2,
                    num_mini_batches=2,
                    app_args={"num_features": 64, "num_parts": 8,
                              "batch_size": 512, "lam": 0.02})
    s = run_job(job, ctx).summary()
    assert s["num_batches"] == 4
    assert s["mse"] < 1.0


def test_gbt_gpu():
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda"))
    job = JobConfig(job_id="g_gbt", app="gbt", max_num_epochs=4,
                    num_mini_batches=2,
                    app_args={"num_features": 16, "batch_size": 2048,
                              "num_bins": 32, "max_depth": 4,
                              "step_size": 0.3})
    s = run_job(job, ctx).summary()
    assert s["num_batches"] == 8
    assert s["mse"] < 6.0


def test_pagerank_gpu_matches_cpu():
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.pregel.engine import PregelEngine
    from harmony_amd.pregel.graphapps import (PageRankComputation,
                                              make_ring_plus_random_graph)
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane
    from harmony_amd.utils import stable_seed

    n = 256
    outs = {}
    for dev in ("cpu", "cuda"):
        ctx = init_executor(RuntimeConfig(device=dev))
        job = JobConfig(job_id=f"g_pr_{dev}", app="pagerank", app_args={})
        cp = ControlPlane(ctx.store, 0, 1)
        comp = PageRankComputation(num_iters=12)
        engine = PregelEngine(job, comp, n, ctx, cp)
        g = make_ring_plus_random_graph(n, 4, 0, n, ctx.device,
                                        stable_seed("gpr", "graph", 0))
        engine.set_graph(g)
        outs[dev] = engine.run().squeeze(1).cpu()
    assert torch.allclose(outs["cpu"], outs["cuda"], atol=1e-5)


def test_migration_gpu_single():
    """Block drop/adopt on device shards (single rank)."""
    from harmony_amd.config import TableConfig
    from harmony_amd.et.table import Table

    cfg = TableConfig(table_id="gm", num_keys=1024, value_dim=8, num_blocks=16)
    t = Table(cfg, 0, 1, torch.device("cuda"))
    keys = torch.arange(1024, device="cuda")
    t.put_local(keys, torch.randn(1024, 8, device="cuda"))
    before = t.get(keys).clone()
    data = t.drop_blocks([3, 7])
    t.adopt_blocks(data)
    assert torch.equal(t.get(keys), before)


def test_checkpoint_gpu_roundtrip():
    from harmony_amd.config import TableConfig
    from harmony_amd.et.checkpoint import CheckpointManager
    from harmony_amd.et.table import Table

    cfg = TableConfig(table_id="gck", num_keys=256, value_dim=8, num_blocks=8)
    t = Table(cfg, 0, 1, torch.device("cuda"))
    t.put_local(torch.arange(256, device="cuda"),
                torch.randn(256, 8, device="cuda"))
    cm = CheckpointManager(temp_root="/tmp/gck_t", commit_root="/tmp/gck_c")
    cm.checkpoint(t, "gapp", "c1")
    t2 = Table(cfg, 0, 1, torch.device("cuda"))
    cm.load_into(t2, "gapp", "c1")
    assert torch.equal(t.shard.cpu(), t2.shard.cpu())


@pytest.mark.gpu
def test_lda_wave_sampler_gpu_converges():
    """K7c wave-per-doc MH sampler: no bit-exact oracle (scheduling-
    dependent interleave) — validate statistically: log-likelihood must
    improve over epochs and land near the serial alias sampler's."""
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cuda"))
    lls = {}
    for sampler in ("alias", "alias_wave"):
        job = JobConfig(job_id=f"lw_{sampler}", app="lda",
                        max_num_epochs=4, num_mini_batches=2,
                        app_args={"num_vocabs": 3000, "num_topics": 64,
                                  "tokens_per_doc": 32,
                                  "docs_per_batch": 1024,
                                  "sampler": sampler})
        s = run_job(job, ctx).summary()
        lls[sampler] = s["log_likelihood"]
    # both negative; at this toy scale the wave sampler's run-to-run
    # variance is larger than at bench scale (measured ~1.4% there) —
    # bound it loosely; the real convergence evidence is
    # profiles/r02_lda_wave.md
    assert lls["alias_wave"] < 0
    assert abs(lls["alias_wave"] - lls["alias"]) < 0.12 * abs(lls["alias"])
