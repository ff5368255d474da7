"""Live block migration between GPU shards — ownership-first.

Reference: et/evaluator/impl/MigrationExecutor.java:48 (ownership flips
BEFORE data moves, per-block OwnershipMsg/Ack then chunked DataMsg) +
driver/impl/MigrationManager.java (orchestration, BlockManager update,
subscriber broadcast).

MI355X shape: migration is a COLLECTIVE executed by every rank of the job at
a quiesced point (a NET-phase ticket between mini-batches — the reference's
per-block read/write locks collapse to phase-level mutual exclusion):
  1. every rank applies the identical ownership flip (ownership-first: any
     access issued after this phase routes to the new owner),
  2. block data moves src->dst as batched RCCL point-to-point sends over
     xGMI (one contiguous buffer per (src,dst) pair, not per block),
  3. receivers adopt into their contiguous shard, senders drop.
No update can be lost: pushes are collective and strictly ordered before or
after the migration phase by the task-unit sequencer.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch
import torch.distributed as dist


def plan_transfers(table, moves: Dict[int, int], rank: int
                   ) -> Tuple[Dict[int, List[int]], Dict[int, List[int]]]:
    """What this rank sends/receives. moves: {block_id: dst_rank}; the data
    holder is the table's LOCAL placement (not the ownership map, which may
    already be flipped)."""
    sends: Dict[int, List[int]] = {}
    recvs: Dict[int, List[int]] = {}
    local = set(table.owned_blocks)
    for b, dst in moves.items():
        src = table.ownership.owner_of_int(b)  # pre-flip owner expected
        if src == dst:
            continue
        if b in local:
            sends.setdefault(dst, []).append(b)
        elif dst == rank:
            recvs.setdefault(src, []).append(b)
    return sends, recvs


def migrate(table, moves: Dict[int, int], rank: int, world_size: int,
            group=None) -> None:
    """Collective: all ranks call with the identical `moves` map."""
    moves = {b: d for b, d in moves.items()
             if table.ownership.owner_of_int(b) != d}
    if not moves:
        return
    if hasattr(table, "migrate_blocks"):
        # one-sided table: its own quiesce/read/remap protocol (the data
        # move is a one-sided peer-HBM read, not RCCL p2p)
        table.migrate_blocks(moves)
        return
    sends, recvs = plan_transfers(table, moves, rank)
    # 1. ownership-first flip (every rank, identically)
    table.ownership.update_many(moves)
    if world_size == 1 or not dist.is_initialized():
        return
    # 2. batched p2p data movement over xGMI. All sends+recvs go in ONE
    # batch_isend_irecv group: individually-enqueued NCCL p2p kernels on a
    # shared comm stream deadlock when two ranks exchange blocks (each
    # rank's send kernel waits for the peer's recv, which is queued behind
    # the peer's own send).
    vdim = table.cfg.value_dim
    bs = table.block_size
    p2p_ops = []
    recv_bufs: Dict[int, Tuple[List[int], torch.Tensor]] = {}
    for src in sorted(recvs):
        blocks = sorted(recvs[src])
        buf = torch.empty((len(blocks) * bs, vdim), dtype=table.dtype,
                          device=table.device)
        recv_bufs[src] = (blocks, buf)
        p2p_ops.append(dist.P2POp(dist.irecv, buf, src, group=group))
    for dst in sorted(sends):
        blocks = sorted(sends[dst])
        buf = torch.cat([table.local_block_view(b) for b in blocks]).contiguous()
        p2p_ops.append(dist.P2POp(dist.isend, buf, dst, group=group))
    if p2p_ops:
        for r in dist.batch_isend_irecv(p2p_ops):
            r.wait()
    # 3. adopt + drop
    all_sent = [b for bs_ in sends.values() for b in bs_]
    if all_sent:
        table.drop_blocks(all_sent)
    adopted: Dict[int, torch.Tensor] = {}
    for src, (blocks, buf) in recv_bufs.items():
        for i, b in enumerate(blocks):
            adopted[b] = buf[i * bs:(i + 1) * bs]
    if adopted:
        table.adopt_blocks(adopted)


def rebalance_moves(table, target_counts: List[int]) -> Dict[int, int]:
    """Greedy pairing of overfull -> underfull ranks (the reference
    HomogeneousOptimizer's priority-queue TransferStep generation,
    HomogeneousOptimizer.java:484-510) — returns a moves map."""
    world = len(target_counts)
    have = {r: list(table.ownership.owned_blocks(r)) for r in range(world)}
    moves: Dict[int, int] = {}
    surplus = [(r, have[r]) for r in range(world)
               if len(have[r]) > target_counts[r]]
    deficit = [r for r in range(world) if len(have[r]) < target_counts[r]]
    di = 0
    for r, blocks in surplus:
        extra = len(blocks) - target_counts[r]
        for b in blocks[-extra:] if extra else []:
            while di < len(deficit):
                d = deficit[di]
                pending = sum(1 for x in moves.values() if x == d)
                if len(have[d]) + pending < target_counts[d]:
                    moves[b] = d
                    break
                di += 1
            if di >= len(deficit):
                break
    return moves
