"""Trainer SPI — the 6-phase per-app contract.

Reference: dolphin/core/worker/Trainer.java:44-92
(initGlobalSettings / setMiniBatchData / pullModel / localCompute /
pushUpdate / onEpochFinished / evaluateModel / cleanup).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import torch


@dataclass
class TrainerContext:
    """Everything a trainer sees of the runtime."""

    job_id: str
    rank: int
    world_size: int
    device: torch.device
    tables: Dict[str, Any] = field(default_factory=dict)   # table_id -> Table
    app_args: Dict[str, Any] = field(default_factory=dict)
    stream: Optional[torch.cuda.Stream] = None

    def table(self, name: str):
        return self.tables[name]


class Trainer:
    """Subclass per app. All tensors live on ctx.device; localCompute runs on
    the job's HIP stream (set by the worker tasklet)."""

    def __init__(self, ctx: TrainerContext):
        self.ctx = ctx

    def initialize(self) -> None:
        """Once per tasklet before the first epoch (initGlobalSettings)."""

    def set_batch_data(self, batch: Any) -> None:
        """Receive the mini-batch (setMiniBatchData)."""
        self.batch = batch

    def pull_model(self) -> None:
        """NET phase: pull needed model rows from the PS tables."""

    def local_compute(self) -> None:
        """COMP phase: the hot HIP kernels."""

    def push_update(self) -> None:
        """NET phase: push deltas to the PS tables."""

    def on_epoch_finished(self, epoch: int) -> None:
        """End-of-epoch hook (loss logging, step-size decay)."""

    def evaluate_model(self) -> Dict[str, float]:
        """Offline model evaluation (reference ModelEvaluator)."""
        return {}

    def cleanup(self) -> None:
        pass

    def num_batch_examples(self) -> int:
        """Examples processed in the current batch (for dataProcessingRate)."""
        b = getattr(self, "batch", None)
        if b is None:
            return 0
        if isinstance(b, torch.Tensor):
            return b.shape[0]
        if hasattr(b, "num_examples"):
            return int(b.num_examples)
        try:
            return len(b)
        except TypeError:
            return 0
