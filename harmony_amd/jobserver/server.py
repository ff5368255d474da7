"""The job server runtime: rank 0 command socket + per-rank job dispatchers.

Reference: jobserver/driver/JobServerDriver.java:56 (ClientMessageHandler
:222-266, shutdown :178-214), JobDispatcher.java:59-85 (per-job thread:
setup tables -> register -> GlobalTaskUnitScheduler.onJobStart -> run),
ResourcePool.java:39-106.

Launch (start_jobserver.sh):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 -m harmony_amd.jobserver.server [flags]

Flow: a client connects to localhost:<port> (default 7008, reference
Parameters.java:29) and sends one JSON line {cmd: SUBMIT|WAIT|STATUS|
SHUTDOWN, ...}. Rank 0 schedules (JobScheduler SPI), publishes the job
record through the control store; every rank's dispatcher processes records
in index order, collectively creates the job's process subgroup (ordered as
an '__admin__' NET phase so group creation cannot interleave with running
jobs' collectives), and members run the job in a tasklet thread.
"""

from __future__ import annotations

import json
import socket
import threading
import time
import traceback
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from harmony_amd.config import DEFAULT_JOBSERVER_PORT, JobConfig, RuntimeConfig
from harmony_amd.jobserver.scheduler import ResourcePool, load_scheduler
from harmony_amd.runtime.bootstrap import ExecutorContext, init_executor
from harmony_amd.runtime.control import ControlPlane, TaskUnitScheduler

ADMIN_JOB = "__admin__"


@dataclass
class JobView:
    """A job-scoped slice of the executor context (job-local rank/world and
    the job's process subgroup)."""

    rank: int
    world_size: int
    device: torch.device
    store: object
    group: object
    global_ranks: List[int]
    backend: str = "sub"

    def new_data_plane(self):
        from harmony_amd.et.comm import DataPlane

        if self.world_size == 1:
            return None
        return DataPlane(self.group, self.rank, self.world_size, self.device)


class JobServerDriver:
    def __init__(self, ctx: ExecutorContext, scheduler: str = "default",
                 port: int = DEFAULT_JOBSERVER_PORT,
                 hb_period: float = 2.0):
        self.ctx = ctx
        self.hb_period = hb_period
        self.cp = ControlPlane(ctx.store, ctx.rank, ctx.world_size)
        # world 1: jobs use local tables (no collectives) — no ordering
        self.tus = TaskUnitScheduler(self.cp, {ADMIN_JOB},
                                     multi_job=ctx.world_size > 1)
        self.tus.set_drawer(ADMIN_JOB, ctx.rank == 0)
        self.pool = ResourcePool(ctx.world_size)
        self.scheduler = load_scheduler(scheduler)
        self.port = port
        self._pending: List[JobConfig] = []
        self._lock = threading.Lock()
        self._threads: List[threading.Thread] = []
        self._listener: Optional[socket.socket] = None

    # ------------------------------------------------------------- rank 0

    def _publish(self, job: JobConfig, ranks: List[int]) -> None:
        idx = self.cp.read("js/njobs")
        rec = {"job": json.loads(job.to_json()), "ranks": ranks}
        self.ctx.store.set(f"js/job/{idx}", json.dumps(rec))
        self.cp.incr("js/njobs", 1)
        self.pool.running[job.job_id] = ranks

    def _try_schedule(self) -> None:
        with self._lock:
            still = []
            for job in self.scheduler.order(list(self._pending)):
                ranks = self.scheduler.on_job_arrival(job, self.pool)
                if ranks:
                    self._publish(job, sorted(set(ranks)))
                else:
                    still.append(job)
            self._pending = still

    def _handle_conn(self, conn: socket.socket) -> None:
        try:
            data = conn.makefile().readline()
            msg = json.loads(data)
            cmd = msg.get("cmd")
            if cmd == "SUBMIT":
                job = JobConfig(**msg["job"])
                with self._lock:
                    self._pending.append(job)
                self._try_schedule()
                conn.sendall((json.dumps({"status": "accepted",
                                          "job_id": job.job_id}) + "\n").encode())
            elif cmd == "WAIT":
                jid = msg["job_id"]
                deadline = time.monotonic() + float(msg.get("timeout", 3600))
                while time.monotonic() < deadline:
                    if self.cp.flag_set(f"js/result/{jid}"):
                        res = self.ctx.store.get(f"js/result/{jid}").decode()
                        conn.sendall((res + "\n").encode())
                        break
                    time.sleep(0.05)
                else:
                    conn.sendall((json.dumps({"status": "timeout"}) + "\n").encode())
            elif cmd == "STATUS":
                conn.sendall((json.dumps({
                    "running": self.pool.running,
                    "pending": [j.job_id for j in self._pending],
                    "world_size": self.ctx.world_size}) + "\n").encode())
            elif cmd == "SHUTDOWN":
                if msg.get("wait_jobs", True):
                    while self.pool.running or self._pending:
                        self._reap_finished()
                        time.sleep(0.05)
                self.cp.set_flag("js/shutdown")
                conn.sendall((json.dumps({"status": "ok"}) + "\n").encode())
        except Exception:  # noqa: BLE001
            traceback.print_exc()
        finally:
            conn.close()

    def _reap_finished(self) -> None:
        # called from the listener loop AND connection-handler threads
        with self._lock:
            done = [jid for jid in list(self.pool.running)
                    if self.cp.flag_set(f"js/result/{jid}")]
            for jid in done:
                self.pool.running.pop(jid, None)
        for jid in done:
            self.scheduler.on_job_finish(jid, self.pool)
        if done:
            self._try_schedule()

    def _listen_loop(self) -> None:
        srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("127.0.0.1", self.port))
        srv.listen(16)
        srv.settimeout(0.2)
        self._listener = srv
        while not self.cp.flag_set("js/shutdown"):
            self._reap_finished()
            try:
                conn, _ = srv.accept()
            except socket.timeout:
                continue
            threading.Thread(target=self._handle_conn, args=(conn,),
                             daemon=True).start()
        srv.close()

    # --------------------------------------------------------- every rank

    def _run_job_thread(self, job: JobConfig, view: JobView) -> None:
        from harmony_amd.dolphin.master import run_job
        from harmony_amd.pregel.runner import PREGEL_APPS, run_pregel_job

        try:
            if job.app in PREGEL_APPS:
                summary = run_pregel_job(job, view, cp=self.cp, tus=self.tus)
            else:
                stream = (torch.cuda.Stream()
                          if view.device.type == "cuda" else None)
                metrics = run_job(job, view, cp=self.cp, tus=self.tus,
                                  stream=stream)
                summary = metrics.summary()
        except Exception as e:  # noqa: BLE001
            traceback.print_exc()
            summary = {"status": "failed", "error": str(e), "rank": view.rank}
        self.ctx.store.set(f"js/result/{job.job_id}/r{view.rank}",
                           json.dumps(summary))
        if view.rank == 0:
            # aggregate per-rank summaries
            agg = []
            for r in range(view.world_size):
                key = f"js/result/{job.job_id}/r{r}"
                self.ctx.store.wait([key])
                agg.append(json.loads(self.ctx.store.get(key)))
            self.ctx.store.set(f"js/result/{job.job_id}", json.dumps({
                "status": ("failed" if any(s.get("status") == "failed"
                                           for s in agg) else "done"),
                "per_rank": agg}))

    def dispatch_loop(self) -> None:
        """Process published job records in index order (every rank)."""
        next_idx = 0
        my_jobs = {ADMIN_JOB}
        while True:
            n = self.cp.read("js/njobs")
            while next_idx < n:
                rec = json.loads(self.ctx.store.get(f"js/job/{next_idx}"))
                job = JobConfig(**rec["job"])
                ranks = rec["ranks"]
                from harmony_amd.utils.joblog import job_logger

                job_logger(job.job_id, self.ctx.rank).info(
                    "scheduled on executors %s", ranks)
                if self.ctx.rank in ranks:
                    my_jobs.add(job.job_id)
                    self.tus.set_jobs(my_jobs)
                    # job-local rank 0 draws this job's NET tickets
                    self.tus.set_drawer(job.job_id,
                                        self.ctx.rank == ranks[0])
                # ordered group creation: an admin NET phase (all ranks)
                with self.tus.net(ADMIN_JOB, next_idx + 1):
                    group = (dist.new_group(ranks)
                             if dist.is_initialized() and
                             len(ranks) < self.ctx.world_size else None)
                if self.ctx.rank in ranks:
                    view = JobView(rank=ranks.index(self.ctx.rank),
                                   world_size=len(ranks),
                                   device=self.ctx.device,
                                   store=self.ctx.store, group=group,
                                   global_ranks=ranks)
                    t = threading.Thread(target=self._run_job_thread,
                                         args=(job, view), daemon=True)
                    t.start()
                    self._threads.append(t)
                next_idx += 1
            if self.cp.flag_set("js/shutdown") and next_idx >= self.cp.read("js/njobs"):
                break
            time.sleep(0.02)
        # fail-fast: cancelled tasklets unwind via JobCancelled raised from
        # control-plane waits, so a short join suffices; a clean shutdown
        # waits for running jobs to finish
        failed = self.cp.flag_set("js/failed")
        for t in self._threads:
            t.join(timeout=15 if failed else 600)

    # ------------------------------------------------- failure detection

    def _heartbeat_loop(self) -> None:
        """Every rank heartbeats; rank 0 watches. Recovery is fail-fast, as
        in the reference (JobServerDriver failed-evaluator handlers throw —
        TODO #677 'no recovery'): a dead executor marks the server failed
        and shuts it down; restart + checkpoint-restore is the story."""
        period = self.hb_period
        while not self.cp.flag_set("js/shutdown"):
            self.cp.store.set(f"js/hb/{self.ctx.rank}", str(time.time()))
            if self.ctx.is_master:
                now = time.time()
                for r in range(self.ctx.world_size):
                    try:
                        last = float(self.ctx.store.get(f"js/hb/{r}"))
                    except Exception:  # noqa: BLE001
                        continue
                    if now - last > 10 * period:
                        print(f"[jobserver] executor {r} heartbeat lost "
                              f"({now - last:.0f}s) — failing fast",
                              flush=True)
                        self.cp.set_flag("js/failed")
                        self.cp.set_flag("js/shutdown")
                        return
            time.sleep(period)

    def run(self) -> None:
        if self.ctx.is_master:
            lt = threading.Thread(target=self._listen_loop, daemon=True)
            lt.start()
        hb = threading.Thread(target=self._heartbeat_loop, daemon=True)
        hb.start()
        self.dispatch_loop()
        # on fail-fast the dead executor can never reach a barrier
        if dist.is_initialized() and not self.cp.flag_set("js/failed"):
            dist.barrier()
        # executor close: commit temp checkpoints (reference ChkpManagerSlave
        # commitAllLocalChkps on close, ChkpManagerSlave.java:226). After the
        # barrier every rank's block files are on disk; rank 0 moves the dirs.
        if self.ctx.is_master and not self.cp.flag_set("js/failed"):
            from harmony_amd.et.checkpoint import commit_all_pending

            n = commit_all_pending()
            if n:
                print(f"[jobserver] committed {n} checkpoint(s) on close",
                      flush=True)


def main() -> None:
    import argparse

    p = argparse.ArgumentParser()
    p.add_argument("-scheduler", "--scheduler", default="default")
    p.add_argument("-port", "--port", type=int, default=DEFAULT_JOBSERVER_PORT)
    p.add_argument("-device", "--device", default="auto")
    args, _ = p.parse_known_args()
    ctx = init_executor(RuntimeConfig(device=args.device))
    JobServerDriver(ctx, scheduler=args.scheduler, port=args.port).run()


if __name__ == "__main__":
    main()
