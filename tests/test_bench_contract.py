"""The driver depends on bench.py's JSON line contract — lock it down."""

import json
import subprocess
import sys


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
