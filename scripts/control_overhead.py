#!/usr/bin/env python3
"""Measure the control-plane tax at N ranks: NET-ticket sequencing + SSP.

Round-1 VERDICT item 2: at N=8 each NET phase cost ~2+ TCPStore RTTs
(compare_set/add ~55 us, blocking get ~110 us loopback) and each SSP poll
scanned all W clock counters. This script measures the per-step control
overhead of the ticket sequencer and the SSP clock at world W with J
co-located jobs, for the round-1 (v1) and round-2 (v2) protocols, with NO
compute or collectives — pure control cost.

Run:  python scripts/control_overhead.py --world 8 --jobs 3 --steps 200
"""

from __future__ import annotations

import argparse
import json
import multiprocessing as mp
import threading
import time

import torch.distributed as dist

from harmony_amd.runtime.control import ControlPlane, SSPClock, TaskUnitScheduler


class V1TaskUnitScheduler(TaskUnitScheduler):
    """Round-1 ticket protocol (one store key per seq, wait+get per foreign
    seq) for A/B comparison."""

    def _ticket(self, job_id, phase_idx, lookahead=0):
        key = f"tu/seq_of/{job_id}/{phase_idx}"  # v1: no batching/lookahead
        token = f"P{self.cp.rank}"
        cur = self.cp.store.compare_set(key, "", token)
        if cur == token.encode():
            seq = self.cp.incr("tu/seq", 1)
            self.cp.store.set(f"tu/job_of/{seq}", job_id)
            self.cp.store.set(key + "/v", str(seq))
            return seq
        self.cp.store.wait([key + "/v"])
        return int(self.cp.store.get(key + "/v"))

    def _job_of(self, seq):
        job = self._job_cache.get(seq)
        if job is None:
            self.cp.store.wait([f"tu/job_of/{seq}"])
            job = self.cp.store.get(f"tu/job_of/{seq}").decode()
            self._job_cache[seq] = job
        return job


class V1SSPClock(SSPClock):
    """Round-1 SSP: store-incremented clocks, O(W) scan per poll."""

    def _ckey(self, r):
        return f"ssp1/{self.job_id}/clock/{r}"

    def tick_and_wait(self, rank, wait=True):
        mine = self.cp.incr(self._ckey(rank), 1)
        if wait and self.slack >= 0 and self.num_workers > 1:
            while True:
                slowest = min(self.cp.read(self._ckey(r))
                              for r in range(self.num_workers))
                if mine - slowest <= self.slack:
                    break
                time.sleep(0.0005)
        return True


def run_rank(rank, world, jobs, steps, version, port, out_q,
             ssp_wait=False, slack=4):
    import faulthandler, os, sys
    if os.environ.get("CO_DEBUG"):
        sys.stderr = open(f"/tmp/co_r{rank}.err", "w", buffering=1)
        faulthandler.dump_traceback_later(int(os.environ["CO_DEBUG"]),
                                          exit=True)
    from harmony_amd.runtime.bootstrap import ThreadLocalTCPStore

    store = ThreadLocalTCPStore("127.0.0.1", port, world,
                                is_master=(rank == 0))
    cp = ControlPlane(store, rank, world)
    job_ids = [f"job{j}" for j in range(jobs)]
    cls = TaskUnitScheduler if version == "v2" else V1TaskUnitScheduler
    tus = cls(cp, set(job_ids), multi_job=jobs > 1)
    for j in job_ids:
        tus.set_drawer(j, rank == 0)
    ssp_cls = SSPClock if version == "v2" else V1SSPClock
    clocks = {j: ssp_cls(cp, j, world, slack=slack) for j in job_ids}

    def job_thread(jid, res):
        import traceback
        phase = [0]

        def nxt():
            phase[0] += 1
            return phase[0]

        t0 = time.perf_counter()
        for _ in range(steps):
            # collective-plane jobs tick without blocking (wait=False); a
            # blocking SSP here can deadlock against the ticket order
            # given >=1 step of cross-rank skew (--deadlock-demo runs the
            # hazardous config; see SSPClock.tick_and_wait)
            clocks[jid].tick_and_wait(rank, wait=ssp_wait)
            with tus.net(jid, nxt(), lookahead=1):
                pass                        # PULL (control only)
            with tus.net(jid, nxt()):
                pass                        # PUSH (control only)
        res[jid] = time.perf_counter() - t0

    def job_thread_safe(jid, res, _f=job_thread):
        try:
            _f(jid, res)
        except BaseException:
            import traceback
            with open(f"/tmp/co_err_r{rank}_{jid}.txt", "w") as f:
                traceback.print_exc(file=f)
            res[jid] = None

    res = {}
    import sys
    def log(msg):
        print(f"[r{rank} {time.strftime('%H:%M:%S')}] {msg}", flush=True)
    log("store up")
    # barrier-ish start
    store.add("start", 1)
    while store.add("start", 0) < world:
        time.sleep(0.001)
    log("start barrier passed")
    t0 = time.perf_counter()
    ts = [threading.Thread(target=job_thread_safe, args=(j, res)) for j in job_ids]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    wall = time.perf_counter() - t0
    log(f"threads done wall={wall:.3f}")
    if any(res.get(j) is None for j in job_ids):
        print(f"rank {rank}: thread died, see /tmp/co_err_r{rank}_*.txt",
              flush=True)
    # finish protocol: rank 0 hosts the store server, so it must exit
    # LAST — non-zero ranks signal and leave (never polling a dying
    # server); rank 0 polls its own server until everyone signalled
    if rank == 0:
        while store.add("fin", 0) < world - 1:
            time.sleep(0.005)
    else:
        store.add("fin", 1)
    out_q.put((rank, wall, res))


def measure(version, world, jobs, steps, port, ssp_wait=False, slack=4):
    q = mp.Queue()
    ps = [mp.Process(target=run_rank,
                     args=(r, world, jobs, steps, version, port, q,
                           ssp_wait, slack))
          for r in range(world)]
    for p in ps:
        p.start()
    outs = [q.get(timeout=600) for _ in ps]
    for p in ps:
        p.join()
    wall = max(o[1] for o in outs)
    return wall


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=8)
    ap.add_argument("--jobs", type=int, default=3)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--versions", default="v1,v2")
    ap.add_argument("--deadlock-demo", action="store_true",
                    help="run with blocking SSP waits inside ticketed jobs "
                         "at slack=0 — the hazardous configuration. The "
                         "SSP/ticket wait cycle needs >=1 step of cross-"
                         "rank skew, so with these empty step bodies the "
                         "hang is TIMING-DEPENDENT (it was observed in "
                         "development, not deterministic here)")
    args = ap.parse_args()
    if args.deadlock_demo:
        print("deadlock demo: blocking SSP + tickets, slack=0 "
              "(timing-dependent hang; run under a timeout)")
        measure("v2", args.world, args.jobs, args.steps, 29660,
                ssp_wait=True, slack=0)
        return
    port0 = 29650
    out = {}
    for i, version in enumerate(args.versions.split(",")):
        wall = measure(version, args.world, args.jobs, args.steps,
                       port0 + i)
        per_step_ms = wall / args.steps * 1000
        out[version] = {"wall_s": round(wall, 3),
                        "control_ms_per_step": round(per_step_ms, 4)}
        print(f"{version}: {args.world} ranks x {args.jobs} jobs x "
              f"{args.steps} steps -> {wall:.3f}s total, "
              f"{per_step_ms:.3f} ms/step control overhead")
    if "v1" in out and "v2" in out:
        sp = (out['v1']['control_ms_per_step']
              / out['v2']['control_ms_per_step'])
        out["speedup_v1_over_v2"] = round(sp, 2)
    print(json.dumps(out))


if __name__ == "__main__":
    mp.set_start_method("spawn", force=True)
    main()
