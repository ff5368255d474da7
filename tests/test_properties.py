"""Property-based tests (hypothesis) for the pure-function core: the
partitioners' routing invariants, the merge algebra of the wire
aggregation, and the GBT tensor codec. These are the functions every
data-plane exchange relies on — a violated invariant here silently
corrupts routing or updates at any world size. References:
HashBasedBlockPartitioner.java:31, OrderingBasedBlockPartitioner.java:30,
CommManager per-block write serialization (merge associativity),
GBTreeListCodec (mlapps/serialization/)."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from harmony_amd.et.partitioner import (HashBasedPartitioner,
                                        OrderingBasedPartitioner)
from harmony_amd.et.update_functions import merge_key_deltas


@settings(max_examples=50, deadline=None)
@given(num_keys=st.integers(1, 10000), num_blocks=st.integers(1, 64),
       seed=st.integers(0, 2**31 - 1))
def test_ordering_partitioner_invariants(num_keys, num_blocks, seed):
    p = OrderingBasedPartitioner(num_keys, num_blocks)
    g = torch.Generator().manual_seed(seed)
    keys = torch.randint(0, num_keys, (256,), generator=g)
    b = p.block_of(keys)
    off = p.offset_in_block(keys)
    # every key maps into a valid block and a valid slot, reversibly
    assert bool((b >= 0).all()) and bool((b < p.num_blocks).all())
    assert bool((off >= 0).all()) and bool((off < p.block_size).all())
    assert bool((b * p.block_size + off == keys).all())
    # scalar path agrees with the vector path
    assert p.block_of_int(int(keys[0])) == int(b[0])
    # key_range covers the keyspace exactly once
    total = sum(len(p.key_range(i)) for i in range(p.num_blocks))
    assert total == num_keys


@settings(max_examples=50, deadline=None)
@given(num_blocks=st.integers(1, 1024),
       keys=st.lists(st.integers(0, 2**62), min_size=1, max_size=128))
def test_hash_partitioner_invariants(num_blocks, keys):
    p = HashBasedPartitioner(num_blocks)
    t = torch.tensor(keys, dtype=torch.int64)
    b = p.block_of(t)
    assert bool((b >= 0).all()) and bool((b < num_blocks).all())
    # deterministic and scalar-consistent
    assert torch.equal(b, p.block_of(t))
    assert p.block_of_int(keys[0]) == int(b[0])


@settings(max_examples=50, deadline=None)
@given(n=st.integers(1, 200), k=st.integers(1, 8),
       mode=st.sampled_from(["add", "assign", "min", "lda_counts"]),
       seed=st.integers(0, 2**31 - 1))
def test_merge_key_deltas_matches_oracle(n, k, mode, seed):
    g = torch.Generator().manual_seed(seed)
    keys = torch.randint(0, max(1, n // 3), (n,), generator=g)
    deltas = torch.randn(n, k, generator=g)
    uniq, merged = merge_key_deltas(keys, deltas, mode)
    assert torch.equal(uniq, torch.unique(keys))
    for i, u in enumerate(uniq.tolist()):
        rows = deltas[keys == u]
        if mode == "min":
            exp = rows.min(dim=0).values
            assert torch.allclose(merged[i], exp)
        elif mode == "assign":
            # "last wins (any is valid)": the merged row must BE one of
            # the duplicate rows, applied whole (no cross-row mixing)
            assert any(torch.equal(merged[i], r) for r in rows)
        else:
            assert torch.allclose(merged[i], rows.sum(dim=0), atol=1e-5)


@settings(max_examples=30, deadline=None)
@given(depth=st.integers(1, 5), n=st.integers(0, 6),
       seed=st.integers(0, 2**31 - 1))
def test_gbt_codec_roundtrip(depth, n, seed):
    from harmony_amd.mlapps.gbt import GBTree, decode_trees, encode_trees

    g = torch.Generator().manual_seed(seed)
    ni = (1 << depth) - 1
    items = []
    for j in range(n):
        items.append((j, GBTree(
            depth=depth,
            feature=torch.randint(0, 50, (ni,), generator=g).tolist(),
            threshold=torch.randint(-1, 30, (ni,), generator=g).tolist(),
            leaf_value=torch.randn(1 << depth, generator=g).tolist())))
    enc = encode_trees(items, depth)
    assert enc.dtype == torch.int32 and enc.shape[0] == n
    dec = decode_trees(enc, depth)
    assert len(dec) == n
    for (k0, t0), (k1, t1) in zip(items, dec):
        assert k0 == k1 and t0.feature == t1.feature
        assert t0.threshold == t1.threshold
        assert torch.allclose(torch.tensor(t0.leaf_value),
                              torch.tensor(t1.leaf_value))


@settings(max_examples=40, deadline=None)
@given(world=st.integers(2, 8), blocks=st.integers(2, 64),
       seed=st.integers(0, 2**31 - 1))
def test_rebalance_moves_reach_targets(world, blocks, seed):
    """rebalance_moves (the reference's priority-queue TransferStep
    pairing): applying the produced moves lands exactly on the target
    counts, and only genuinely-moving blocks appear."""
    import torch

    from harmony_amd.config import TableConfig
    from harmony_amd.et.migration import rebalance_moves
    from harmony_amd.et.table import Table

    g = torch.Generator().manual_seed(seed)
    cfg = TableConfig(table_id="rb", num_keys=blocks * 4, value_dim=2,
                      num_blocks=blocks, update_fn="add", init_fn="zeros")
    t = Table(cfg, 0, world, torch.device("cpu"))
    # random-but-valid target: a permutation of the current counts
    cur = [len(t.ownership.owned_blocks(r)) for r in range(world)]
    perm = torch.randperm(world, generator=g).tolist()
    target = [cur[p] for p in perm]
    moves = rebalance_moves(t, target)
    after = list(cur)
    for b, dst in moves.items():
        src = t.ownership.owner_of_int(b)
        assert src != dst            # no vacuous moves
        after[src] -= 1
        after[dst] += 1
    assert after == target, (cur, target, moves)


@settings(max_examples=40, deadline=None)
@given(blocks=st.integers(1, 128), world=st.integers(1, 9),
       seed=st.integers(0, 2**31 - 1))
def test_ownership_invariants(blocks, world, seed):
    """Ownership map invariants (reference BlockManager even split +
    collective update): initial partition is contiguous/even within 1,
    every block always has exactly one owner, counts are conserved, the
    version advances on every mutation, and slot maps are dense."""
    import torch

    from harmony_amd.et.ownership import Ownership

    o = Ownership(blocks, world)
    cnt = o.counts()
    assert sum(cnt) == blocks
    assert max(cnt) - min(cnt) <= 1           # even within one block
    # contiguous: owner tensor is non-decreasing
    assert bool((o.owner[1:] >= o.owner[:-1]).all())
    g = torch.Generator().manual_seed(seed)
    v0 = o.version
    moves = {int(torch.randint(0, blocks, (1,), generator=g)):
             int(torch.randint(0, world, (1,), generator=g))
             for _ in range(min(8, blocks))}
    o.update_many(moves)
    assert o.version == v0 + 1
    for b, r in moves.items():
        assert o.owner_of_int(b) == r          # last write wins per key
    assert sum(o.counts()) == blocks           # conservation
    for r in range(world):
        slots = o.slot_of(r)
        ob = o.owned_blocks(r)
        assert sorted(slots.keys()) == ob      # dense, sorted slot map
        assert sorted(slots.values()) == list(range(len(ob)))


@settings(max_examples=40, deadline=None)
@given(seed=st.integers(0, 2**31 - 1), nops=st.integers(0, 8))
def test_plan_json_roundtrip(seed, nops):
    """Plans are published through the control store as JSON (collective
    and async-queue delivery): serialization must be lossless for every
    op type, deps, and the benefit estimate."""
    import random

    from harmony_amd.optimizer.plan import (DropTableOp, MoveOp, Plan,
                                            SetBatchShareOp, StartWorkerOp,
                                            StopWorkerOp)

    rng = random.Random(seed)
    ops = []
    for _ in range(nops):
        kind = rng.randrange(5)
        if kind == 0:
            ops.append(MoveOp(f"t{rng.randrange(3)}",
                              tuple((rng.randrange(64), rng.randrange(8))
                                    for _ in range(rng.randrange(4)))))
        elif kind == 1:
            ops.append(SetBatchShareOp(tuple((r, rng.randrange(1, 100))
                                             for r in range(rng.randrange(1, 4)))))
        elif kind == 2:
            ops.append(StopWorkerOp(rng.randrange(8)))
        elif kind == 3:
            ops.append(StartWorkerOp(rng.randrange(8), rng.randrange(1, 50)))
        else:
            ops.append(DropTableOp(f"t{rng.randrange(3)}"))
    deps = [(i, j) for i in range(len(ops)) for j in range(i + 1, len(ops))
            if rng.random() < 0.3]
    p = Plan(ops=ops, deps=deps, estimated_benefit=rng.random())
    q = Plan.from_json(p.to_json())
    assert q.ops == p.ops
    assert [tuple(d) for d in q.deps] == [tuple(d) for d in p.deps]
    assert abs(q.estimated_benefit - p.estimated_benefit) < 1e-12
