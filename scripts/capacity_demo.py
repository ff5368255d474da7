#!/usr/bin/env python3
"""HBM3E capacity demonstration: scale the model tables toward the
MI355X's 288 GB and show per-step time stays roofline-bound (the design
keeps tensors resident — bigger shards, same step shape).

Run on a GPU box: python scripts/capacity_demo.py"""
import json
import sys
import time

sys.path.insert(0, ".")

import torch  # noqa: E402

from harmony_amd import mlapps  # noqa: E402
from harmony_amd.config import JobConfig, RuntimeConfig  # noqa: E402
from harmony_amd.runtime.bootstrap import init_executor  # noqa: E402
from harmony_amd.runtime.control import ControlPlane  # noqa: E402


def run_one(app, ctx, cp, steps=6, **app_args):
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    job = JobConfig(job_id=f"cap_{app}_{app_args.get('num_vocabs', app_args.get('num_cols'))}",
                    app=app, num_mini_batches=2, num_worker_blocks=2,
                    app_args=app_args)
    t0 = time.monotonic()
    tables, tr, pr = mlapps.get_app(app).build(job, ctx, cp)
    if hasattr(tr, "initialize"):
        tr.initialize()
    torch.cuda.synchronize()
    build_s = time.monotonic() - t0

    def step(i):
        b = pr.blocks[i % len(pr.blocks)]
        tr.set_batch_data(b)
        tr.pull_model()
        tr.local_compute()
        tr.push_update()

    step(0)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for i in range(steps):
        step(i)
    torch.cuda.synchronize()
    ms = (time.monotonic() - t0) / steps * 1e3
    free, total = torch.cuda.mem_get_info()
    out = {"app": app, "args": {k: v for k, v in app_args.items()
                                if isinstance(v, int)},
           "build_s": round(build_s, 1), "ms_per_step": round(ms, 3),
           "alloc_GB": round(torch.cuda.max_memory_allocated() / 2**30, 1),
           "hbm_used_GB": round((total - free) / 2**30, 1)}
    print(json.dumps(out), flush=True)
    for t in tables.values():
        if hasattr(t, "drop_blocks"):
            try:
                t.drop_blocks(list(t.owned_blocks))
            except Exception:
                pass
    del tables, tr, pr
    torch.cuda.empty_cache()
    return out


def main():
    ctx = init_executor(RuntimeConfig(device="cuda"))
    cp = ControlPlane(ctx.store, 0, 1)
    for vocab in (1 << 20, 1 << 22, 1 << 24):
        run_one("lda", ctx, cp, num_vocabs=vocab, num_topics=256,
                num_docs=65536, tokens_per_doc=128, docs_per_batch=16384,
                sampler="alias_wave")
    for cols in (1 << 22, 1 << 24, 1 << 26):
        run_one("nmf", ctx, cp, num_rows=32768, num_cols=cols, rank=100,
                nnz_per_row=128, batch_size=16384)


if __name__ == "__main__":
    main()
