"""Model accessor: the worker's view of the PS model table.

Reference: dolphin/core/worker/ModelAccessor.java (pull/push SPI) with
ETModelAccessor (direct table access, ETModelAccessor.java:43) and
CachedModelAccessor (worker-side cache with background refresh,
CachedModelAccessor.java:40-130).

MI355X shape: pull = collective gather (keys or all), push = collective
scatter of deltas; the cached accessor keeps the last pulled rows resident in
HBM and refreshes them every `refresh_batches` batches instead of per batch —
the device-memory analogue of the reference's Guava cache + refresh thread.
"""

from __future__ import annotations

import time
from typing import Dict, Optional

import torch

METRIC_PULL_TIME = "total_pull_time_sec"
METRIC_PUSH_TIME = "total_push_time_sec"


class ETModelAccessor:
    def __init__(self, table):
        self.table = table
        self.metrics: Dict[str, float] = {METRIC_PULL_TIME: 0.0, METRIC_PUSH_TIME: 0.0}

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        t0 = time.perf_counter()
        out = self.table.get(keys)
        self.metrics[METRIC_PULL_TIME] += time.perf_counter() - t0
        return out

    def pull_all(self) -> torch.Tensor:
        t0 = time.perf_counter()
        out = self.table.pull_all()
        self.metrics[METRIC_PULL_TIME] += time.perf_counter() - t0
        return out

    def push(self, keys: torch.Tensor, deltas: torch.Tensor,
             assume_unique: bool = False) -> None:
        t0 = time.perf_counter()
        self.table.update(keys, deltas, assume_unique=assume_unique)
        self.metrics[METRIC_PUSH_TIME] += time.perf_counter() - t0

    def push_dense(self, grad_full: torch.Tensor) -> None:
        t0 = time.perf_counter()
        self.table.push_dense(grad_full)
        self.metrics[METRIC_PUSH_TIME] += time.perf_counter() - t0

    def get_and_reset_metrics(self) -> Dict[str, float]:
        m = dict(self.metrics)
        for k in self.metrics:
            self.metrics[k] = 0.0
        return m


class CachedModelAccessor(ETModelAccessor):
    """Keeps the full model resident; re-pulls every `refresh_batches` pulls.
    Pushes are applied write-through to the cache with the table's update
    function (reference CachedModelAccessor pushes via UpdateFunction)."""

    def __init__(self, table, refresh_batches: int = 1):
        super().__init__(table)
        self.refresh_batches = max(1, refresh_batches)
        self._cache: Optional[torch.Tensor] = None
        self._age = 0

    def pull_all(self) -> torch.Tensor:
        if self._cache is None or self._age >= self.refresh_batches:
            # clone: the local-mode pull returns the live shard, and a cache
            # aliasing the table would see (and double-apply) every update
            self._cache = super().pull_all().clone()
            self._age = 0
        self._age += 1
        return self._cache

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        return self.pull_all()[keys]

    def push(self, keys: torch.Tensor, deltas: torch.Tensor,
             assume_unique: bool = False) -> None:
        super().push(keys, deltas, assume_unique=assume_unique)
        if self._cache is not None:
            from harmony_amd.et import update_functions as uf

            fn = uf.update_fn(self.table.cfg.update_fn)
            rows = self._cache[keys]
            self._cache[keys] = fn(rows, deltas.to(rows.dtype),
                                   **self.table.cfg.update_args)


class CachedOneSidedAccessor:
    """Cached accessor with a BACKGROUND refresh thread (reference
    CachedModelAccessor.java:40-130: a Guava cache re-pulled every
    MODEL_REFRESH_SEC by a refresh thread). Background pulls are only safe
    on the ASYNC plane — a one-sided table's pull is a gather kernel, so a
    timer thread cannot desynchronize any collective order. Pushes go
    through immediately and are applied write-through to the cache."""

    def __init__(self, table, refresh_sec: float = 0.5):
        import threading

        self.table = table
        self.refresh_sec = refresh_sec
        self._cache: Optional[torch.Tensor] = None
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self.refreshes = 0
        self._thread = threading.Thread(target=self._refresh_loop,
                                        daemon=True)
        self._thread.start()

    def _refresh_loop(self) -> None:
        while not self._stop.wait(self.refresh_sec):
            fresh = self.table.pull_full()
            with self._lock:
                self._cache = fresh
                self.refreshes += 1

    def close(self) -> None:
        self._stop.set()
        self._thread.join(timeout=5)

    def pull_all(self) -> torch.Tensor:
        with self._lock:
            if self._cache is None:
                self._cache = self.table.pull_full()
            return self._cache

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        return self.pull_all()[keys]

    def push(self, keys, deltas, assume_unique: bool = False) -> None:
        self.table.push(keys, deltas)
        from harmony_amd.et import update_functions as uf

        with self._lock:
            if self._cache is not None:
                fn = uf.update_fn(self.table.cfg.update_fn)
                rows = self._cache[keys]
                self._cache[keys] = fn(rows, deltas.to(rows.dtype),
                                       **self.table.cfg.update_args)

    def drain(self):
        if hasattr(self.table, "drain"):
            return self.table.drain()
        return 0


class OneSidedAccessor:
    """ETModelAccessor-shaped facade over an et.onesided.OneSidedTable:
    pulls and pushes are direct xGMI kernels, never collectives, so a
    worker using it steps at its own pace (true async PS; SSP slack is
    the only cross-worker coupling). Reference: workers talking to remote
    tablets through the async op queue (RemoteAccessOpSender)."""

    def __init__(self, table):
        import collections

        self.table = table
        self._keys = None
        self.metrics = collections.defaultdict(float)

    def pull_all(self):
        return self.table.pull_full()

    def pull(self, keys):
        return self.table.pull(keys)

    def drain(self):
        # owner-side apply-queue drain (one-sided v2 rings; no-op for
        # add-algebra tables) — called once per batch after push
        if hasattr(self.table, "drain"):
            return self.table.drain()
        return 0

    def push_dense(self, grad_full):
        if self._keys is None or self._keys.shape[0] != grad_full.shape[0]:
            import torch as _t

            self._keys = _t.arange(grad_full.shape[0],
                                   device=grad_full.device)
        self.table.push(self._keys, grad_full)

    def push(self, keys, deltas, assume_unique=False):
        self.table.push(keys, deltas)
