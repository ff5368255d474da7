#!/usr/bin/env python3
"""Flagship benchmark: 3 concurrent PS jobs (NMF + MLR + LDA) sharing N MI355X.

This measures the BASELINE.json headline metric — aggregate examples/sec
(the reference's dataProcessingRate, WorkerTasklet.java:203) for three
concurrent parameter-server jobs co-scheduled on the same GPUs, weak scaling
(per-GPU batch work fixed as N grows).

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
  #        --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

One "step" = one mini-batch of EACH of the three jobs on every rank (the
jobs run concurrently in tasklet threads on per-job HIP streams; NET phases
are globally ticket-ordered). Data is synthetic, weights random-init.
"""

from __future__ import annotations

import argparse
import json
import os
import threading
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--apps", type=str, default="nmf,mlr,lda")
    p.add_argument("--device", type=str, default="auto")
    p.add_argument("--mode", type=str, default="direct",
                   choices=["direct", "runtime"],
                   help="direct: the driver-contract timed loop. runtime: "
                        "the SAME 3 jobs through the real per-job runtime "
                        "(run_job -> WorkerTasklet: SSP clock, global "
                        "barriers, metric collection) — the delta vs "
                        "direct is the control-plane cost users pay "
                        "(VERDICT r01 item 5)")
    # per-GPU shapes (weak scaling: these are PER RANK)
    p.add_argument("--nmf-rank", type=int, default=100)
    p.add_argument("--nmf-cols", type=int, default=65536)
    p.add_argument("--nmf-rows-per-batch", type=int, default=16384)
    p.add_argument("--nmf-nnz-per-row", type=int, default=128)
    p.add_argument("--mlr-classes", type=int, default=10)
    p.add_argument("--mlr-features", type=int, default=16384)
    p.add_argument("--mlr-batch", type=int, default=16384)
    p.add_argument("--lda-vocab", type=int, default=100000)
    p.add_argument("--lda-topics", type=int, default=256)
    p.add_argument("--lda-docs-per-batch", type=int, default=16384)
    p.add_argument("--lda-tokens-per-doc", type=int, default=128)
    p.add_argument("--lda-alias-refresh", type=int, default=4)
    p.add_argument("--lda-sampler", type=str, default="alias_wave",
                   choices=["exact", "alias", "alias_wave"])
    p.add_argument("--cu-partition", type=str, default="",
                   help="direct mode: pin each job's stream to a disjoint "
                        "CU range, e.g. 'lda=96,nmf=96,mlr=64' (multi-"
                        "tenant partitioning via hipExtStreamCreateWith"
                        "CUMask; empty = shared full chip)")
    p.add_argument("--elastic", action="store_true",
                   help="runtime mode only: mid-run StopWorker + live "
                        "block migration of the last rank, then StartWorker"
                        " + migration back (BASELINE config #5); emits an "
                        "elastic_timeline field")
    return p.parse_args()


def make_jobs(args, world: int):
    from harmony_amd.config import JobConfig

    n_blocks = 4  # resident synthetic blocks per rank, cycled
    # LDA keeps 8 resident blocks: at the default 16384 docs/batch that is
    # 131k docs per rank -> the BASELINE "LDA 1M-doc/100k-vocab" config at
    # 8 GPUs (weak scaling)
    lda_blocks = 8
    jobs = {}
    if "nmf" in args.apps:
        jobs["nmf"] = JobConfig(
            job_id="bench_nmf", app="nmf", num_mini_batches=n_blocks,
            num_worker_blocks=n_blocks,
            app_args={"num_cols": args.nmf_cols, "rank": args.nmf_rank,
                      "nnz_per_row": args.nmf_nnz_per_row,
                      "rows_per_batch": args.nmf_rows_per_batch,
                      "step_size": 0.01})
    if "mlr" in args.apps:
        jobs["mlr"] = JobConfig(
            job_id="bench_mlr", app="mlr", num_mini_batches=n_blocks,
            num_worker_blocks=n_blocks,
            app_args={"num_classes": args.mlr_classes,
                      "num_features": args.mlr_features,
                      "num_parts_per_class": 8,
                      "batch_size": args.mlr_batch, "step_size": 0.01})
    if "lda" in args.apps:
        jobs["lda"] = JobConfig(
            job_id="bench_lda", app="lda", num_mini_batches=lda_blocks,
            num_worker_blocks=lda_blocks,
            app_args={"num_vocabs": args.lda_vocab,
                      "num_topics": args.lda_topics,
                      "tokens_per_doc": args.lda_tokens_per_doc,
                      "docs_per_batch": args.lda_docs_per_batch,
                      "sampler": args.lda_sampler,
                      "alias_refresh": args.lda_alias_refresh})
    return jobs


class JobBench:
    """One job's tasklet: builds tables/trainer/data, steps on command."""

    def __init__(self, job, ctx, cp, tus, use_stream: bool):
        from harmony_amd import mlapps

        self.job = job
        self.tus = tus
        app = mlapps.get_app(job.app)
        self.tables, self.trainer, self.provider = app.build(job, ctx, cp)
        self.stream = (torch.cuda.Stream() if use_stream else None)
        self.blocks = self.provider.blocks
        self._phase = 0
        self._i = 0
        self.examples_per_batch = 0

    def _next_phase(self):
        self._phase += 1
        return self._phase

    def initialize(self):
        import contextlib

        sctx = (torch.cuda.stream(self.stream) if self.stream is not None
                else contextlib.nullcontext())
        with sctx:
            with self.tus.net(self.job.job_id, self._next_phase()):
                self.trainer.initialize()

    def step(self):
        import contextlib

        sctx = (torch.cuda.stream(self.stream) if self.stream is not None
                else contextlib.nullcontext())
        with sctx:
            batch = self.blocks[self._i % len(self.blocks)]
            self._i += 1
            self.trainer.set_batch_data(batch)
            # PULL draws PUSH's ticket too (one store round-trip per step)
            with self.tus.net(self.job.job_id, self._next_phase(),
                              lookahead=1):
                self.trainer.pull_model()
            self.trainer.local_compute()
            with self.tus.net(self.job.job_id, self._next_phase()):
                self.trainer.push_update()
            self.examples_per_batch = self.trainer.num_batch_examples()


def run_runtime_mode(args, ctx, cp, jobs):
    """Drive the SAME jobs through the real runtime: run_job ->
    WorkerTasklet (SSP tick, global barriers, per-batch metrics). Reports
    each job's steady-state rate (epochs after the first; epoch 0 carries
    one-time inits: alias tables, route caches) and the aggregate.
    Reference loop this exercises: dolphin/core/worker/WorkerTasklet.java:96-168."""
    import dataclasses

    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.control import TaskUnitScheduler

    rank, world = ctx.rank, ctx.world_size
    dev_cuda = ctx.device.type == "cuda"
    shaped = {}
    for name, job in jobs.items():
        nb = job.num_mini_batches
        epochs = 1 + max(1, -(-args.steps // nb))   # 1 warmup + timed
        shaped[name] = dataclasses.replace(
            job, job_id=job.job_id + "_rt", max_num_epochs=epochs,
            optimizer_period=(max(2, args.steps // 5)
                              if args.elastic and name == "nmf" else 8),
            trace_path=(os.environ.get("HARMONY_BENCH_TRACE", "") + name
                        if os.environ.get("HARMONY_BENCH_TRACE") else None))
    tus = TaskUnitScheduler(cp, {j.job_id for j in shaped.values()},
                            multi_job=len(shaped) > 1 and world > 1)
    for j in shaped.values():
        tus.set_drawer(j.job_id, rank == 0)
    results = {}

    def runner(name, job):
        stream = torch.cuda.Stream() if dev_cuda else None
        opt = None
        if args.elastic and name == "nmf" and world > 1:
            from harmony_amd.optimizer.optimizers import SampleOptimizers

            opt = SampleOptimizers.elastic_showcase(world - 1)
        results[name] = run_job(job, ctx, cp=cp, tus=tus, stream=stream,
                                optimizer=opt)

    if dev_cuda:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    threads = [threading.Thread(target=runner, args=(n, j))
               for n, j in shaped.items()]
    t0 = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    if dev_cuda:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    makespan = time.perf_counter() - t0

    per_job = {}
    total_rate = 0.0
    steps_total = 0
    for name, mc in results.items():
        eps = [e for e in mc.epochs if e.epoch_idx >= 1]
        ex = sum(e.num_examples for e in eps)
        tsec = sum(e.epoch_time_sec for e in eps)
        steady = [b for b in mc.batches if b.epoch_idx >= 1]
        net_ms = (sum(b.net_wait_sec for b in steady)
                  / max(1, len(steady)) * 1e3)
        bt_ms = (sum(b.batch_time_sec for b in steady)
                 / max(1, len(steady)) * 1e3)
        # MAX time over ranks, SUM examples over ranks
        if dist.is_initialized():
            tt = torch.tensor([tsec], dtype=torch.float64,
                              device=ctx.device if dev_cuda else "cpu")
            dist.all_reduce(tt, op=dist.ReduceOp.MAX)
            tsec = float(tt.cpu())
            et = torch.tensor([float(ex)], dtype=torch.float64,
                              device=ctx.device if dev_cuda else "cpu")
            dist.all_reduce(et, op=dist.ReduceOp.SUM)
            ex = float(et.cpu())
        rate = ex / tsec if tsec else 0.0
        nb = jobs[name].num_mini_batches
        per_job[name] = {"examples_per_sec": rate,
                         "timed_epochs": len(eps),
                         "ms_per_batch": tsec / max(1, len(eps) * nb) * 1e3,
                         # measured control-plane cost:
                         # NET-ticket sequencer wait per batch and its
                         # share of the batch wall time
                         "net_wait_ms": net_ms,
                         "net_wait_pct": (100.0 * net_ms / bt_ms
                                          if bt_ms else 0.0)}
        total_rate += rate
        steps_total += len(eps) * nb
    timeline = None
    if args.elastic and "nmf" in results:
        print(f"[elastic] rank{rank} applied_plans="
              f"{getattr(results['nmf'], '_applied_plans', '?')}",
              file=__import__('sys').stderr)
    if args.elastic and "nmf" in results and dist.is_initialized():
        mine = [(b.epoch_idx, b.batch_idx,
                 round(b.batch_time_sec * 1e3, 2), b.num_examples)
                for b in results["nmf"].batches]
        allt = [None] * world
        dist.all_gather_object(allt, mine)
        timeline = allt
    if rank == 0:
        ctl = {j: {"draw_s": round(tus.stat_draw_s.get(j, 0.0), 3),
                   "order_s": round(tus.stat_order_s.get(j, 0.0), 3)}
               for j in sorted(tus.stat_draw_s)}
        if os.environ.get("HARMONY_DUMP_ORDER"):
            seqs = sorted(tus._job_cache)
            print("ORDER:", "".join(tus._job_cache[s][6] for s in seqs),
                  file=__import__("sys").stderr)
        out = {
            "metric": "aggregate_examples_per_sec_3job_runtime",
            "rank0_control_split": ctl,
            "elastic_timeline": timeline,
            "value": total_rate,
            "unit": "examples/s",
            "n_gpus": world,
            "steps": steps_total,
            "warmup": 0,
            "ms_per_step": (makespan / max(1, steps_total)) * 1e3,
            "makespan_sec": makespan,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "note": "full runtime path (WorkerTasklet: SSP+barriers+"
                    "metrics); steady-state epochs (>=1); compare with "
                    "--mode direct for the control-plane delta",
            "per_job": per_job,
            "config": {"model": "+".join(jobs.keys()) +
                       " concurrent PS jobs (runtime path)",
                       "global_batch": None, "seq_len": None,
                       "parallelism": f"ps-dp{world}"},
        }
        print(json.dumps(out))


def main():
    if os.environ.get("HARMONY_DUMP_STACKS"):
        import faulthandler

        rank = os.environ.get("RANK", "0")
        global _stackf
        _stackf = open(f"/tmp/bench_stacks_r{rank}.txt", "w", buffering=1)
        faulthandler.dump_traceback_later(
            int(os.environ["HARMONY_DUMP_STACKS"]), exit=False,
            file=_stackf, repeat=True)
    args = parse_args()
    from harmony_amd.config import RuntimeConfig
    from harmony_amd.runtime.bootstrap import init_executor
    from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

    ctx = init_executor(RuntimeConfig(device=args.device))
    rank, world = ctx.rank, ctx.world_size
    dev_cuda = ctx.device.type == "cuda"
    cp = ControlPlane(ctx.store, rank, world)
    jobs = make_jobs(args, world)
    if args.mode == "runtime":
        run_runtime_mode(args, ctx, cp, jobs)
        return
    # tickets order CROSS-RANK collectives; at world 1 there are none, so
    # a torchrun-launched N=1 (TCPStore present) must not pay ~0.1 ms/phase
    # of store round-trips for nothing
    multi = len(jobs) > 1 and world > 1
    tus = TaskUnitScheduler(cp, {j.job_id for j in jobs.values()},
                            multi_job=multi)
    for j in jobs.values():
        tus.set_drawer(j.job_id, rank == 0)
    cu_streams = None
    if args.cu_partition and dev_cuda:
        from harmony_amd.utils.custreams import cu_partitioned_streams

        shares = {}
        for part in args.cu_partition.split(","):
            name, n = part.split("=")
            shares[name.strip()] = int(n)
        cu_streams = cu_partitioned_streams(shares)
    benches = []
    for name, j in jobs.items():
        b = JobBench(j, ctx, cp, tus, use_stream=dev_cuda)
        if cu_streams is not None and name in cu_streams:
            b.stream = cu_streams[name]
        benches.append(b)
    for b in benches:
        b.initialize()

    def run_steps(n: int):
        errs = []

        def worker(b):
            try:
                for _ in range(n):
                    b.step()
            except Exception:  # noqa: BLE001
                import traceback

                errs.append(traceback.format_exc())

        threads = [threading.Thread(target=worker, args=(b,)) for b in benches]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        if errs:
            raise RuntimeError(errs[0])

    def sync_all():
        if dev_cuda:
            torch.cuda.synchronize()
        if dist.is_initialized():
            dist.barrier()
        if dev_cuda:
            torch.cuda.synchronize()

    # warmup: at least the requested W steps, and keep repeating W-step
    # rounds until the GPU has been busy ~2s — MI355X SCLK takes O(1s) of
    # sustained load to ramp, so a handful of ~1ms steps would time the
    # first K steps at low clocks. Timed region is untouched (full work).
    warm_min_s = float(os.environ.get("HARMONY_BENCH_MIN_WARMUP_S", "2.0"))
    run_steps(args.warmup)
    if dev_cuda and args.warmup > 0:
        sync_all()
        tw = time.perf_counter()
        while True:
            # all ranks must agree on extra rounds (run_steps is
            # collective at N>1): MAX-reduce the local continue flag
            cont = time.perf_counter() - tw < warm_min_s
            if dist.is_initialized():
                t = torch.tensor([1.0 if cont else 0.0],
                                 device=ctx.device if dev_cuda else "cpu")
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
                cont = bool(t.item() > 0)
            if not cont:
                break
            run_steps(args.warmup)
            torch.cuda.synchronize()
    sync_all()
    t0 = time.perf_counter()
    run_steps(args.steps)
    sync_all()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=ctx.device if dev_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu())

    examples_per_step_rank = sum(b.examples_per_batch for b in benches)
    total_examples = examples_per_step_rank * args.steps * world
    value = total_examples / elapsed
    if rank == 0:
        out = {
            "metric": "aggregate_examples_per_sec_3job",
            "value": value,
            "unit": "examples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "makespan_sec": elapsed,   # BASELINE metric: + makespan
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "+".join(jobs.keys()) + " concurrent PS jobs",
                "global_batch": examples_per_step_rank * world,
                "seq_len": None,
                "parallelism": f"ps-dp{world}",
                "nmf": {"rank": args.nmf_rank, "cols": args.nmf_cols,
                        "rows_per_batch": args.nmf_rows_per_batch,
                        "nnz_per_row": args.nmf_nnz_per_row},
                "mlr": {"classes": args.mlr_classes,
                        "features": args.mlr_features,
                        "batch": args.mlr_batch},
                "lda": {"vocab": args.lda_vocab, "topics": args.lda_topics,
                        "docs_per_batch": args.lda_docs_per_batch,
                        "tokens_per_doc": args.lda_tokens_per_doc,
                        "sampler": args.lda_sampler},
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
