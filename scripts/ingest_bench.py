#!/usr/bin/env python3
"""Large-input ingestion at 8 ranks: generate a
multi-GB sample_nmf-format file once, then measure exactly-N split + native
parse throughput with 8 parallel reader processes (the jobserver's loading
path: dataloader.parse_nmf_split -> textparse.cpp)."""
import os
import sys
import time

PATH = "/tmp/harmony_ingest_nmf.txt"


def gen(target_gb=2.0):
    import numpy as np

    rng = np.random.default_rng(0)
    row = 0
    t0 = time.perf_counter()
    with open(PATH, "w") as f:
        while os.path.getsize(PATH) < target_gb * (1 << 30) if False else \
                f.tell() < target_gb * (1 << 30):
            cols = rng.integers(0, 100000, size=64)
            vals = rng.random(64)
            f.write(f"{row}: " + " ".join(
                f"{c},{v:.4f}" for c, v in zip(cols, vals)) + "\n")
            row += 1
    print(f"generated {os.path.getsize(PATH)/1e9:.2f} GB, {row} rows "
          f"in {time.perf_counter()-t0:.1f}s")


def worker(rank, world):
    from harmony_amd import dataloader as dl

    t0 = time.perf_counter()
    rows, cols, vals = dl.parse_nmf_split(PATH, rank, world)
    dt = time.perf_counter() - t0
    return (dt, int(vals.numel()))


def main():
    if not os.path.exists(PATH) or os.path.getsize(PATH) < 1 << 30:
        gen(2.0)
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from tests.dist_helper import run_dist

    size = os.path.getsize(PATH)
    res = run_dist(worker, world=8, timeout=600)
    wall = max(r[0] for r in res)
    nnz = sum(r[1] for r in res)
    print(f"8-rank ingest of {size/1e9:.2f} GB: {wall:.2f} s wall "
          f"(max rank), {size/1e9/wall:.2f} GB/s aggregate, "
          f"{nnz/1e6:.1f}M nonzeros parsed")


if __name__ == "__main__":
    main()
