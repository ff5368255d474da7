#!/usr/bin/env python3
"""One-command verification mirroring the driver's checks.

CPU box (no GPU):   python scripts/verify_all.py
GPU box (gpurun):   python scripts/verify_all.py --gpu

Runs, in order: build (hipcc cross-compile + import), the CPU test suite,
and with --gpu additionally the GPU suite, smoke(), and a short bench with
a JSON-contract check. Exits nonzero on the first failure.

Budget note: each step is a separate python process, and the FIRST torch
import on a fresh box can take 1-2 min while the image pages in — budget
>=10 min of box time for --gpu (a 3.5-min clamp killed the first run)."""
import argparse
import json
import subprocess
import sys
import time

REQUIRED_BENCH_FIELDS = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]


def step(name, cmd, timeout):
    t0 = time.monotonic()
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
    dt = time.monotonic() - t0
    ok = r.returncode == 0
    print(f"[{'ok' if ok else 'FAIL':4s}] {name:28s} {dt:6.1f}s")
    if not ok:
        print(r.stdout[-2000:])
        print(r.stderr[-2000:])
        sys.exit(1)
    return r


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpu", action="store_true")
    a = ap.parse_args()
    py = sys.executable
    step("build (graft entry)", [py, "-c",
         "import __graft_entry__ as g; g.build()"], 1800)
    step("cpu tests", [py, "-m", "pytest", "tests/", "-x", "-q",
                       "-m", "not gpu", "-p", "no:cacheprovider"], 1800)
    if not a.gpu:
        print("CPU verification complete (pass --gpu on a GPU box).")
        return
    step("gpu tests", [py, "-m", "pytest", "tests/", "-x", "-q",
                       "-m", "gpu", "-p", "no:cacheprovider"], 1200)
    step("smoke", [py, "-c",
         "import __graft_entry__ as g; g.smoke()"], 300)
    r = step("bench (driver contract)", [py, "bench.py"], 600)
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    missing = [k for k in REQUIRED_BENCH_FIELDS if k not in d]
    assert not missing, f"bench JSON missing fields: {missing}"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    print(f"bench: {d['value']/1e6:.2f}M {d['unit']} "
          f"({d['ms_per_step']:.3f} ms/step)")
    print("GPU verification complete.")


if __name__ == "__main__":
    main()
