"""Block partitioners: key -> block id.

Reference: et/evaluator/impl/{OrderingBasedBlockPartitioner.java:30-74,
HashBasedBlockPartitioner.java:31-55}. The ordering-based partitioner maps
contiguous key ranges to blocks (and exposes the range of a block, used for
block-local key generation and block=mini-batch iteration); the hash-based
partitioner spreads arbitrary keys.

MI355X-first: partitioning is vectorized over torch int64 key tensors so the
routing of a whole mini-batch's keys is one device op, not a per-key hash.
"""

from __future__ import annotations

import torch


class OrderingBasedPartitioner:
    """Contiguous equal-size key ranges; block b owns [b*bs, (b+1)*bs)."""

    def __init__(self, num_keys: int, num_blocks: int):
        assert num_blocks >= 1
        self.num_keys = num_keys
        self.num_blocks = num_blocks
        self.block_size = (num_keys + num_blocks - 1) // num_blocks

    def block_of(self, keys: torch.Tensor) -> torch.Tensor:
        return keys // self.block_size

    def block_of_int(self, key: int) -> int:
        return key // self.block_size

    def key_range(self, block_id: int) -> range:
        """Key range owned by a block (clipped to the real keyspace)."""
        lo = block_id * self.block_size
        hi = min(lo + self.block_size, self.num_keys)
        return range(lo, hi)

    def offset_in_block(self, keys: torch.Tensor) -> torch.Tensor:
        return keys % self.block_size


class HashBasedPartitioner:
    """Mixing hash of the int64 key -> block (for unordered keyspaces)."""

    def __init__(self, num_blocks: int):
        assert num_blocks >= 1
        self.num_blocks = num_blocks

    def block_of(self, keys: torch.Tensor) -> torch.Tensor:
        # splitmix64-style mix, vectorized; work in int64 with wraparound.
        x = keys.to(torch.int64)
        x = x ^ (x >> 30)
        x = x * (-0x61c8864680b583eb)  # 0x9E3779B97F4A7C15 as signed
        x = x ^ (x >> 27)
        x = x * (-0x7ee3623a03d3db3b)  # 0x94D049BB133111EB as signed
        x = x ^ (x >> 31)
        return (x & 0x7FFFFFFFFFFFFFFF) % self.num_blocks

    def block_of_int(self, key: int) -> int:
        return int(self.block_of(torch.tensor([key], dtype=torch.int64))[0])
