"""hipGraph capture for static per-block compute phases.

The COMP phase of a PS mini-batch is a fixed kernel sequence over static
tensors (the batch block is device-resident for the whole job; the pulled
model lands in a preallocated buffer). Capturing it as a hipGraph
(torch.cuda.CUDAGraph == hipGraph on ROCm) replaces 10-20 Python-driven
kernel launches per step with one graph launch — the launch-bound host path
is what co-located jobs contend on.

Usage (inside a trainer):
    self._graphs = GraphRunner()
    ...
    def local_compute(self):
        self._graphs.run(("comp", self._block_idx, self._epoch_key()),
                         self._compute_body)

Rules for the captured body: pure device work — no host syncs, no
collectives, no data-dependent host control flow, inputs/outputs through
tensors that outlive the graph. Capture failures fall back to eager
permanently (logged once). Graphs must be invalidated (clear()) when shard
pointers change (table migration).
"""

from __future__ import annotations

import logging
from typing import Callable, Dict, Hashable

import torch

logger = logging.getLogger("harmony.graphs")


class GraphRunner:
    def __init__(self, enabled: bool = True, warmup_iters: int = 2):
        self.enabled = enabled and torch.cuda.is_available()
        self.warmup_iters = warmup_iters
        self._graphs: Dict[Hashable, torch.cuda.CUDAGraph] = {}
        self._failed = False
        self._stream = torch.cuda.Stream() if self.enabled else None

    def run(self, key: Hashable, body: Callable[[], None],
            state: tuple = ()) -> None:
        """Replay the graph for `key`, capturing it on first use.

        state: tensors the body mutates in place. They are snapshotted
        before warmup and restored before the first real replay, so the
        extra warmup executions leave no trace (stream capture itself
        executes nothing)."""
        if not self.enabled or self._failed:
            body()
            return
        g = self._graphs.get(key)
        if g is not None:
            g.replay()
            return
        try:
            saved = [t.clone() for t in state]
            # warmup on a side stream (allocator + lazy-init kernels)
            self._stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._stream):
                for _ in range(self.warmup_iters):
                    body()
            torch.cuda.current_stream().wait_stream(self._stream)
            g = torch.cuda.CUDAGraph()
            # thread_local: co-located jobs' threads keep allocating while
            # this one captures — the default "global" error mode hard-
            # aborts the whole process on any cross-thread CUDA malloc
            with torch.cuda.graph(g, capture_error_mode="thread_local"):
                body()
            for t, s in zip(state, saved):
                t.copy_(s)
            self._graphs[key] = g
            g.replay()
        except Exception as e:  # noqa: BLE001 — fall back to eager forever
            logger.warning("hipGraph capture failed (%s); eager fallback", e)
            self._failed = True
            self._graphs.clear()
            body()

    def clear(self) -> None:
        """Drop captured graphs (call after migration: shard storage moved)."""
        self._graphs.clear()
