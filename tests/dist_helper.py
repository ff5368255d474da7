"""Spawn N-rank test workers (gloo on CPU here; same code path as RCCL on GPU)."""

from __future__ import annotations

import multiprocessing as mp
import os
import pickle
import socket
import traceback


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(fn, rank, world, port, q, args):
    import sys

    # tee stderr to a per-rank file so run_dist can show it on failure
    err_path = f"/tmp/dist_helper_r{rank}_p{port}.err"
    sys.stderr = open(err_path, "w", buffering=1)
    if os.environ.get("DIST_HELPER_DUMP"):
        import faulthandler

        faulthandler.dump_traceback_later(
            int(os.environ["DIST_HELPER_DUMP"]), exit=False,
            file=sys.stderr, repeat=False)
    try:
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        out = fn(rank, world, *args)
        q.put((rank, "ok", pickle.dumps(out)))
    except Exception:  # noqa: BLE001
        q.put((rank, "err", traceback.format_exc()))
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            try:
                # keep the rank-0 control store alive until everyone is done
                dist.barrier()
            except Exception:  # noqa: BLE001
                pass
            dist.destroy_process_group()


def _stderr_tails(world: int, port: int, n: int = 40) -> str:
    out = []
    for r in range(world):
        p = f"/tmp/dist_helper_r{r}_p{port}.err"
        try:
            lines = open(p).readlines()[-n:]
            out.append(f"--- rank {r} stderr tail ---\n" + "".join(lines))
        except OSError:
            pass
    return "\n".join(out)


def run_dist(fn, world: int = 2, args=(), timeout: int = 120):
    """Run fn(rank, world, *args) in `world` processes; returns [out_rank0..].
    Retries once on INFRASTRUCTURE failures only (port collisions, socket
    resets, spawn stalls) — genuine assertion/logic failures propagate on
    the first attempt."""
    _INFRA = ("EADDRINUSE", "Connection reset", "Connection refused",
              "did not report within", "worker died",
              "Socket Timeout", "wait timeout after")
    try:
        return _run_dist_once(fn, world, args, timeout)
    except (RuntimeError, TimeoutError) as e:
        msg = str(e)
        if "AssertionError" in msg or not any(t in msg for t in _INFRA):
            raise
        return _run_dist_once(fn, world, args, timeout)


def _run_dist_once(fn, world: int = 2, args=(), timeout: int = 120):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = free_port()
    procs = [ctx.Process(target=_entry, args=(fn, r, world, port, q, args))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    import queue as queue_mod
    import time

    try:
        deadline = time.monotonic() + timeout
        while len(results) < world:
            try:
                rank, status, payload = q.get(timeout=2)
            except queue_mod.Empty:
                dead = [(i, p.exitcode) for i, p in enumerate(procs)
                        if not p.is_alive() and p.exitcode != 0]
                if dead:
                    raise RuntimeError(f"worker died: (rank, exitcode)={dead}"
                                       f"\n{_stderr_tails(world, port)}")
                if time.monotonic() > deadline:
                    raise TimeoutError(
                        f"ranks {sorted(set(range(world)) - set(results))} "
                        f"did not report within {timeout}s\n"
                        f"{_stderr_tails(world, port)}")
                continue
            if status == "err":
                raise RuntimeError(f"rank {rank} failed:\n{payload}")
            results[rank] = pickle.loads(payload)
    finally:
        for p in procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
    return [results[r] for r in range(world)]
