"""Training data provider: one block = one mini-batch.

Reference: dolphin/core/worker/ETTrainingDataProvider.java:38 — an epoch
iterates the local blocks of the input table; each block is one mini-batch,
shuffled within the block. Here a "block" is an app-defined device-resident
batch object (dense tensor, CSR triple, token arrays ...) prepared once by
the app's synthetic generator or data parser and kept in HBM for the whole
job — no host<->device staging inside the training loop.
"""

from __future__ import annotations

import random
from typing import Any, Callable, Iterator, List, Optional


class TrainingDataProvider:
    def __init__(self, local_blocks: List[Any], shuffle: bool = True,
                 seed: int = 0,
                 reslice: Optional[Callable[[Any, float], Any]] = None):
        self.blocks = local_blocks
        self.shuffle = shuffle
        self.seed = seed
        # SetBatchShareOp consumption: apps that can re-slice a batch to a
        # fraction of its examples register `reslice(batch, frac)`; the
        # orchestrator's plan then shrinks a slow rank's per-batch work
        # (the number of BATCHES stays identical on every rank — collective
        # counts must match — only each batch's example count changes).
        self.reslice = reslice
        self.share = 1.0
        self._sliced: dict = {}

    def set_share(self, frac: float) -> None:
        """frac in [0, 1]: serve that fraction of each block's examples.
        frac == 0 (StopWorkerOp) serves truly EMPTY batches — the stopped
        worker keeps entering collectives (it still serves its table
        blocks, the reference's worker->server switch) but contributes
        zero keys, shedding its pull/push fan-in."""
        frac = min(1.0, max(frac, 0.0))
        if frac != self.share:
            self.share = frac
            self._sliced.clear()

    @property
    def num_batches(self) -> int:
        return len(self.blocks)

    def _get(self, i: int) -> Any:
        if self.share >= 1.0 or self.reslice is None:
            return self.blocks[i]
        if i not in self._sliced:
            self._sliced[i] = self.reslice(self.blocks[i], self.share)
        return self._sliced[i]

    def epoch_iter(self, epoch: int) -> Iterator[Any]:
        order = list(range(len(self.blocks)))
        if self.shuffle:
            random.Random(self.seed * 100003 + epoch).shuffle(order)
        for i in order:
            yield self._get(i)
