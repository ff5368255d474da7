"""Executor bootstrap: one process per GPU, RCCL data plane + control store.

Replaces the reference's REEF driver/evaluator lifecycle (ExecutorManager,
evalmanager — reference et/driver/impl/ExecutorManager.java:62,
evalmanager/api/EvaluatorManager.java:39). Launch model:

  torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 <prog>

Each rank pins one MI355X, joins the default process group (backend "nccl" ==
RCCL on ROCm; "gloo" for CPU tests), and connects to the control TCPStore
hosted by rank 0.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

from harmony_amd.config import RuntimeConfig

_STORE_KEEPALIVE: list = []


@dataclass
class ExecutorContext:
    rank: int
    world_size: int
    device: torch.device
    backend: str               # "nccl" | "gloo" | "none"
    store: object              # control-plane store (TCPStore or LocalStore)

    @property
    def is_master(self) -> bool:
        return self.rank == 0

    def new_data_plane(self, group=None):
        from harmony_amd.et.comm import DataPlane

        if self.backend == "none":
            return None
        return DataPlane(group, self.rank, self.world_size, self.device)

    def barrier(self):
        if dist.is_initialized():
            dist.barrier()


def _pick_device(cfg_device: str, local_rank: int) -> torch.device:
    if cfg_device == "cpu":
        return torch.device("cpu")
    if cfg_device == "cuda" or (cfg_device == "auto" and torch.cuda.is_available()):
        if not torch.cuda.is_available():
            raise RuntimeError("device=cuda requested but no GPU is visible")
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        return torch.device("cuda", local_rank % torch.cuda.device_count())
    return torch.device("cpu")


def init_executor(cfg: Optional[RuntimeConfig] = None) -> ExecutorContext:
    cfg = cfg or RuntimeConfig()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(cfg.world_size)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    master_addr = os.environ.get("MASTER_ADDR", cfg.master_addr)
    master_port = int(os.environ.get("MASTER_PORT", str(cfg.master_port)))

    device = _pick_device(cfg.device, local_rank)

    if world > 1 or "RANK" in os.environ:
        backend = cfg.backend
        if backend == "auto":
            backend = "nccl" if device.type == "cuda" else "gloo"
        if not dist.is_initialized():
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world,
                timeout=datetime.timedelta(seconds=300))
        store = ThreadLocalTCPStore(master_addr, master_port + 7, world,
                                    is_master=(rank == 0))
        # Pin the store for the process lifetime: if rank 0's TCPStore object
        # is GC'd when the caller's frame dies, the server socket closes and
        # every peer still polling it crashes.
        _STORE_KEEPALIVE.append(store)
    else:
        backend = "none"
        store = LocalStore()

    return ExecutorContext(rank=rank, world_size=world, device=device,
                           backend=backend, store=store)


class ThreadLocalTCPStore:
    """One TCPStore CLIENT per thread.

    A single TCPStore client multiplexes one socket; a thread blocked in
    wait() holds the connection while another thread's compare_set on the
    same client wedges behind it — with concurrent job tasklets this
    deadlocks (observed: one tasklet in wait(tu/job_of/..), the other stuck
    inside compare_set forever). Each thread therefore gets its own client
    connection; the rank-0 master listener lives on the first instance.
    """

    def __init__(self, host: str, port: int, world: int, is_master: bool):
        import threading

        self._host = host
        self._port = port
        self._world = world
        self._tls = threading.local()
        self._lock = threading.Lock()
        self._master_store = None
        self._master_claimed = False
        if is_master:
            self._master_store = self._new_client(is_master=True)

    def _new_client(self, is_master: bool = False):
        from torch.distributed import TCPStore

        # wait_for_workers=False: clients are created lazily per thread, so
        # nobody can count on a fixed connection census.
        return TCPStore(self._host, self._port, self._world,
                        is_master=is_master,
                        timeout=datetime.timedelta(seconds=300),
                        wait_for_workers=False)

    def _client(self):
        c = getattr(self._tls, "c", None)
        if c is None:
            with self._lock:
                if self._master_store is not None and not self._master_claimed:
                    # exactly ONE thread may reuse the master instance — the
                    # claim flag must be process-wide, not thread-local
                    self._master_claimed = True
                    c = self._master_store
            if c is None:
                c = self._new_client()
            self._tls.c = c
        return c

    def set(self, key, value):
        return self._client().set(key, value)

    def get(self, key):
        return self._client().get(key)

    def add(self, key, amount):
        return self._client().add(key, amount)

    def compare_set(self, key, expected, desired):
        return self._client().compare_set(key, expected, desired)

    def append(self, key, value):
        return self._client().append(key, value)

    def wait(self, keys):
        return self._client().wait(keys)

    def check(self, keys):
        return self._client().check(keys)

    def delete_key(self, key):
        return self._client().delete_key(key)

    def multi_get(self, keys):
        return self._client().multi_get(keys)

    def multi_set(self, keys, values):
        return self._client().multi_set(keys, values)


class LocalStore:
    """In-process stand-in for TCPStore in single-process mode (tests, local
    runtime). Implements the subset of the Store API the control plane uses."""

    def __init__(self):
        import threading

        self._d = {}
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)

    def set(self, key: str, value) -> None:
        with self._cv:
            self._d[key] = _to_bytes(value)
            self._cv.notify_all()

    def get(self, key: str) -> bytes:
        import time

        deadline = time.monotonic() + 300
        with self._cv:
            while key not in self._d:
                if not self._cv.wait(timeout=1.0) and time.monotonic() > deadline:
                    raise KeyError(key)
            return self._d[key]

    def add(self, key: str, amount: int) -> int:
        with self._cv:
            cur = int(self._d.get(key, b"0")) + amount
            self._d[key] = str(cur).encode()
            self._cv.notify_all()
            return cur

    def compare_set(self, key: str, expected: str, desired: str) -> bytes:
        with self._cv:
            cur = self._d.get(key)
            if (cur is None and expected == "") or cur == _to_bytes(expected):
                self._d[key] = _to_bytes(desired)
                self._cv.notify_all()
                return _to_bytes(desired)
            return cur if cur is not None else _to_bytes(expected)

    def append(self, key: str, value) -> None:
        with self._cv:
            self._d[key] = self._d.get(key, b"") + _to_bytes(value)
            self._cv.notify_all()

    def wait(self, keys) -> None:
        for k in keys:
            self.get(k)

    def check(self, keys) -> bool:
        with self._lock:
            return all(k in self._d for k in keys)

    def delete_key(self, key: str) -> bool:
        with self._cv:
            return self._d.pop(key, None) is not None

    def multi_get(self, keys):
        return [self.get(k) for k in keys]

    def multi_set(self, keys, values) -> None:
        with self._cv:
            for k, v in zip(keys, values):
                self._d[k] = _to_bytes(v)
            self._cv.notify_all()


def _to_bytes(v) -> bytes:
    if isinstance(v, bytes):
        return v
    return str(v).encode()
