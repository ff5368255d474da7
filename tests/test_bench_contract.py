"""The driver depends on bench.py's JSON line contract — lock it down."""

import json
import subprocess
import sys
from pathlib import Path

REPO = str(Path(__file__).resolve().parents[1])


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--nmf-cols", "512", "--nmf-rows-per-batch", "64",
         "--nmf-nnz-per-row", "8", "--mlr-features", "64",
         "--mlr-batch", "64", "--lda-vocab", "500",
         "--lda-docs-per-batch", "32", "--lda-tokens-per-doc", "8"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"] == "aggregate_examples_per_sec_3job"
    assert isinstance(d["value"], float) and d["value"] > 0
    assert d["unit"] == "examples/s"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["ms_per_step"] > 0 and d["makespan_sec"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["vs_baseline"] is None          # no published baseline
    assert d["dtype"] == "fp32" and d["data"] == "synthetic"
    cfg = d["config"]
    assert "nmf" in cfg and "mlr" in cfg and "lda" in cfg
    assert cfg["parallelism"] == "ps-dp1"
    assert cfg["global_batch"] > 0
    # value is the WHOLE-JOB aggregate: examples per step x steps / elapsed
    assert abs(d["value"] * d["makespan_sec"]
               - cfg["global_batch"] * d["steps"]) < 1e-3 * d["value"]


def test_bench_runtime_mode_contract():
    """--mode runtime drives the same jobs through run_job/WorkerTasklet
    (benchmark through the real runtime)."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--mode", "runtime", "--steps", "2",
         "--nmf-cols", "512", "--nmf-rows-per-batch", "64",
         "--nmf-nnz-per-row", "8", "--mlr-features", "64",
         "--mlr-batch", "64", "--lda-vocab", "500",
         "--lda-docs-per-batch", "32", "--lda-tokens-per-doc", "8"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"] == "aggregate_examples_per_sec_3job_runtime"
    assert d["value"] > 0
    for job in ("nmf", "mlr", "lda"):
        pj = d["per_job"][job]
        assert pj["examples_per_sec"] > 0 and pj["timed_epochs"] >= 1


def test_bench_elastic_contract():
    """--mode runtime --elastic at world 2: the timeline shows the
    stopped rank's zero-example window and recovery (BASELINE config #5
    through the bench contract)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29734", "bench.py", "--mode", "runtime",
         "--elastic", "--gpus", "2", "--steps", "40", "--device", "cpu",
         "--apps", "nmf,mlr", "--nmf-cols", "2048",
         "--nmf-rows-per-batch", "256", "--nmf-nnz-per-row", "8",
         "--mlr-features", "512", "--mlr-batch", "256"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if '"metric"' in l][-1]
    d = json.loads(line)
    tl = d["elastic_timeline"]
    assert tl is not None and len(tl) == 2
    ex1 = [row[3] for row in tl[1]]
    assert 0 in ex1, ex1[:8]            # rank 1 truly stopped mid-run
    assert ex1[0] > 0 and ex1[-1] > 0   # worked at both ends (restarted)


def test_bench_direct_world2_torchrun():
    """The driver's SCALE invocation shape: direct mode under torchrun at
    world 2 (tickets + collectives + clock-warm loop all engaged)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29734", "bench.py", "--gpus", "2",
         "--device", "cpu", "--steps", "3", "--warmup", "1",
         "--nmf-rows-per-batch", "64", "--nmf-cols", "256",
         "--mlr-batch", "64", "--mlr-features", "64",
         "--lda-docs-per-batch", "32", "--lda-vocab", "500",
         "--lda-tokens-per-doc", "8"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(out.stdout.strip().splitlines()[-1])
    assert d["metric"] == "aggregate_examples_per_sec_3job"
    assert d["n_gpus"] == 2 and d["value"] > 0
