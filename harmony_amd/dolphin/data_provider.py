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
from typing import Any, Iterator, List


class TrainingDataProvider:
    def __init__(self, local_blocks: List[Any], shuffle: bool = True, seed: int = 0):
        self.blocks = local_blocks
        self.shuffle = shuffle
        self.seed = seed

    @property
    def num_batches(self) -> int:
        return len(self.blocks)

    def epoch_iter(self, epoch: int) -> Iterator[Any]:
        order = list(range(len(self.blocks)))
        if self.shuffle:
            random.Random(self.seed * 100003 + epoch).shuffle(order)
        for i in order:
            yield self.blocks[i]
