#!/usr/bin/env python3
"""QoS A/B for CU-partitioned co-scheduling: a latency-sensitive MLR
tenant steps continuously while an LDA tenant hammers the GPU. Compare
the MLR tenant's per-step latency distribution with (a) shared full-chip
streams vs (b) MLR pinned to its own 64-CU partition.

Run on a GPU box: python scripts/qos_ab.py"""
import sys
import threading
import time

sys.path.insert(0, ".")

import torch  # noqa: E402

from harmony_amd import mlapps  # noqa: E402
from harmony_amd.config import JobConfig, RuntimeConfig  # noqa: E402
from harmony_amd.runtime.bootstrap import init_executor  # noqa: E402
from harmony_amd.runtime.control import ControlPlane  # noqa: E402


def build(app, ctx, cp, **app_args):
    job = JobConfig(job_id=f"qos_{app}", app=app, num_mini_batches=4,
                    num_worker_blocks=4, app_args=app_args)
    tables, tr, pr = mlapps.get_app(app).build(job, ctx, cp)
    if hasattr(tr, "initialize"):
        tr.initialize()
    torch.cuda.synchronize()
    return tables, tr, pr


def main():
    ctx = init_executor(RuntimeConfig(device="cuda"))
    cp = ControlPlane(ctx.store, 0, 1)
    _, lda_tr, lda_pr = build("lda", ctx, cp, num_vocabs=100000,
                              num_topics=256, tokens_per_doc=128,
                              docs_per_batch=16384, sampler="alias_wave")
    _, mlr_tr, mlr_pr = build("mlr", ctx, cp, num_classes=10,
                              num_features=4096, num_parts_per_class=8,
                              batch_size=4096)

    def step(tr, pr, i):
        b = pr.blocks[i % len(pr.blocks)]
        tr.set_batch_data(b)
        tr.pull_model()
        tr.local_compute()
        tr.push_update()

    def run_mode(mlr_stream, lda_stream, label, secs=4.0):
        stop = threading.Event()
        hammer_n = [0]
        hammer_err = []

        def hammer():
            try:
                with torch.cuda.stream(lda_stream):
                    while not stop.is_set():
                        step(lda_tr, lda_pr, hammer_n[0])
                        hammer_n[0] += 1
                    torch.cuda.synchronize()
            except Exception as e:              # noqa: BLE001
                hammer_err.append(repr(e))

        lat = []
        t = threading.Thread(target=hammer)
        t.start()
        time.sleep(0.5)
        with torch.cuda.stream(mlr_stream):
            end = time.monotonic() + secs
            i = 0
            while time.monotonic() < end:
                t0 = time.perf_counter()
                step(mlr_tr, mlr_pr, i)
                mlr_stream.synchronize()
                lat.append(time.perf_counter() - t0)
                i += 1
        stop.set()
        t.join()
        torch.cuda.synchronize()
        assert not hammer_err, f"co-tenant crashed: {hammer_err[0]}"
        assert hammer_n[0] > 100, f"co-tenant barely ran: {hammer_n[0]}"
        lat.sort()
        n = len(lat)
        p = lambda q: lat[min(n - 1, int(q * n))] * 1e3  # noqa: E731
        print(f"{label}: n={n} lda_steps={hammer_n[0]} p50={p(.5):.3f} "
              f"p95={p(.95):.3f} p99={p(.99):.3f} max={lat[-1]*1e3:.3f} ms")

    # warm EVERYTHING untimed (Tensile kernel selection, alias builds,
    # SCLK ramp) so mode ordering cannot masquerade as isolation
    tw = time.monotonic()
    i = 0
    while time.monotonic() - tw < 2.0:
        step(lda_tr, lda_pr, i)
        step(mlr_tr, mlr_pr, i)
        i += 1
    torch.cuda.synchronize()

    from harmony_amd.utils.custreams import cu_partitioned_streams

    st = cu_partitioned_streams({"mlr": 64, "lda": 192})
    s_sh_m, s_sh_l = torch.cuda.Stream(), torch.cuda.Stream()
    # interleave the modes to cancel any residual drift
    run_mode(s_sh_m, s_sh_l, "shared   ")
    run_mode(st["mlr"], st["lda"], "partition")
    run_mode(s_sh_m, s_sh_l, "shared2  ")
    run_mode(st["mlr"], st["lda"], "partition2")


if __name__ == "__main__":
    main()
