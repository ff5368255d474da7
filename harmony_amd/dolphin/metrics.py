"""Worker/server metrics (reference: dolphin/metric + metrics.avsc).

The headline metric is dataProcessingRate = examples / batch elapsed sec
(reference dolphin/core/worker/WorkerTasklet.java:203). Per-batch and
per-epoch records mirror the reference's BatchMetrics/EpochMetrics fields.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List


@dataclass
class BatchMetrics:
    epoch_idx: int
    batch_idx: int
    num_examples: int
    batch_time_sec: float
    pull_time_sec: float
    comp_time_sec: float
    push_time_sec: float
    net_wait_sec: float = 0.0   # NET-ticket sequencer wait (control plane)

    @property
    def data_processing_rate(self) -> float:
        return self.num_examples / self.batch_time_sec if self.batch_time_sec else 0.0


@dataclass
class EpochMetrics:
    epoch_idx: int
    num_examples: int
    epoch_time_sec: float
    custom: Dict[str, float] = field(default_factory=dict)


class MetricCollector:
    """Per-tasklet metric sink (reference et/metric/MetricCollector.java:14)."""

    def __init__(self, job_id: str, rank: int):
        self.job_id = job_id
        self.rank = rank
        self.batches: List[BatchMetrics] = []
        self.epochs: List[EpochMetrics] = []
        self.custom: Dict[str, float] = {}

    def add_batch(self, m: BatchMetrics) -> None:
        self.batches.append(m)

    def add_epoch(self, m: EpochMetrics) -> None:
        self.epochs.append(m)

    def add_custom(self, key: str, value: float) -> None:
        self.custom[key] = value

    def summary(self) -> Dict[str, float]:
        total_examples = sum(b.num_examples for b in self.batches)
        total_time = sum(b.batch_time_sec for b in self.batches)
        return {
            "job_id": self.job_id,
            "rank": self.rank,
            "num_batches": len(self.batches),
            "total_examples": total_examples,
            "total_batch_time_sec": total_time,
            "data_processing_rate": total_examples / total_time if total_time else 0.0,
            "pull_time_sec": sum(b.pull_time_sec for b in self.batches),
            "comp_time_sec": sum(b.comp_time_sec for b in self.batches),
            "push_time_sec": sum(b.push_time_sec for b in self.batches),
            **self.custom,
        }


class Timer:
    """Pull/push/comp timer (reference dolphin/metric/Tracer.java:28)."""

    def __init__(self):
        self.t0 = 0.0

    def start(self) -> None:
        self.t0 = time.perf_counter()

    def stop(self) -> float:
        return time.perf_counter() - self.t0
