"""Span tracing (reference: utils/trace/HTrace.java:30 — HTrace spans wired
through Tang, serialized across processes via traceinfo.avsc; plus the
lighter dolphin/metric/Tracer.java pull/push/comp timers).

MI355X shape: spans are (name, t0, t1, rank, job, parent) records kept
in-process and optionally flushed to a JSONL file; on GPU, each span also
emits a rocTX range (torch.cuda.nvtx maps to rocTX on ROCm) so rocprofv3
--marker-trace correlates host spans with kernels. Cross-process
parent-span propagation travels in the job record (the control-plane
analogue of the reference's TraceInfo-in-Avro-message)."""

from __future__ import annotations

import json
import os
import threading
import time
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import List, Optional

import torch

_ROCTX = torch.cuda.is_available()


@dataclass
class Span:
    name: str
    t0: float
    t1: float = 0.0
    rank: int = 0
    job: str = ""
    parent: str = ""


class Tracer:
    def __init__(self, rank: int = 0, job: str = "", out_path: Optional[str] = None,
                 sample_every: int = 1):
        self.rank = rank
        self.job = job
        self.out_path = out_path
        self.sample_every = max(1, sample_every)
        self.spans: List[Span] = []
        self._n = 0
        self._lock = threading.Lock()
        self._tls = threading.local()

    @contextmanager
    def span(self, name: str):
        self._n += 1
        sampled = (self._n % self.sample_every) == 0
        parent = getattr(self._tls, "cur", "")
        if sampled:
            if _ROCTX:
                torch.cuda.nvtx.range_push(f"{self.job}/{name}")
            s = Span(name=name, t0=time.perf_counter(), rank=self.rank,
                     job=self.job, parent=parent)
            self._tls.cur = name
        try:
            yield
        finally:
            if sampled:
                s.t1 = time.perf_counter()
                self._tls.cur = parent
                if _ROCTX:
                    torch.cuda.nvtx.range_pop()
                with self._lock:
                    self.spans.append(s)

    def flush(self) -> None:
        if not self.out_path:
            return
        with self._lock, open(self.out_path, "a") as f:
            for s in self.spans:
                f.write(json.dumps({"name": s.name, "t0": s.t0, "t1": s.t1,
                                    "dur_ms": (s.t1 - s.t0) * 1e3,
                                    "rank": s.rank, "job": s.job,
                                    "parent": s.parent}) + "\n")
            self.spans.clear()


_NULL = None


def null_tracer() -> Tracer:
    global _NULL
    if _NULL is None:
        _NULL = Tracer(sample_every=1 << 30)
    return _NULL
