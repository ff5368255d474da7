"""Block ownership: blockId -> owner rank.

Reference: driver-side BlockManager.java:16 (authoritative map, even initial
partition) + evaluator-side OwnershipCache.java:51 (replicated map consulted
on every access). In the SPMD rebuild every rank holds the full map as a
device-friendly int32 tensor; it is updated collectively at migration points
(quiesced by the control plane), so the per-block read-write locks of the
reference collapse to phase-level mutual exclusion (see et/migration.py).
"""

from __future__ import annotations

from typing import Dict, List

import torch


class Ownership:
    def __init__(self, num_blocks: int, world_size: int):
        self.num_blocks = num_blocks
        self.world_size = world_size
        # Even contiguous initial partition (reference BlockManager even split):
        # rank r owns blocks [r*B/W, (r+1)*B/W).
        bounds = [(r * num_blocks) // world_size for r in range(world_size + 1)]
        owner = torch.empty(num_blocks, dtype=torch.int32)
        for r in range(world_size):
            owner[bounds[r]:bounds[r + 1]] = r
        self.owner = owner  # int32 [num_blocks], host tensor (small)
        self._version = 0

    # -- queries ----------------------------------------------------------

    def owner_of(self, block_ids: torch.Tensor) -> torch.Tensor:
        return self.owner[block_ids]

    def owner_of_int(self, block_id: int) -> int:
        return int(self.owner[block_id])

    def owned_blocks(self, rank: int) -> List[int]:
        """Sorted block ids owned by `rank` (defines the shard layout)."""
        return torch.nonzero(self.owner == rank, as_tuple=False).flatten().tolist()

    def num_owned(self, rank: int) -> int:
        return int((self.owner == rank).sum())

    def counts(self) -> List[int]:
        return [self.num_owned(r) for r in range(self.world_size)]

    @property
    def version(self) -> int:
        return self._version

    # -- mutation (collective: all ranks must apply identically) -----------

    def update(self, block_id: int, new_owner: int) -> None:
        self.owner[block_id] = new_owner
        self._version += 1

    def update_many(self, moves: Dict[int, int]) -> None:
        for b, r in moves.items():
            self.owner[b] = r
        self._version += 1

    def slot_of(self, rank: int) -> Dict[int, int]:
        """block id -> local slot index in rank's contiguous shard."""
        return {b: i for i, b in enumerate(self.owned_blocks(rank))}
