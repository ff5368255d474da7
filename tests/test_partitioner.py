import torch

from harmony_amd.et.partitioner import HashBasedPartitioner, OrderingBasedPartitioner


def test_ordering_partitioner_ranges():
    p = OrderingBasedPartitioner(num_keys=100, num_blocks=8)
    assert p.block_size == 13
    keys = torch.arange(100)
    blocks = p.block_of(keys)
    assert int(blocks.min()) == 0
    assert int(blocks.max()) == 7
    # every key is inside its block's range
    for b in range(8):
        r = p.key_range(b)
        for k in r:
            assert p.block_of_int(k) == b
    # offsets round-trip
    offs = p.offset_in_block(keys)
    assert torch.equal(blocks * p.block_size + offs, keys)


def test_hash_partitioner_spread_and_determinism():
    p = HashBasedPartitioner(num_blocks=64)
    keys = torch.arange(10000, dtype=torch.int64)
    b1 = p.block_of(keys)
    b2 = p.block_of(keys)
    assert torch.equal(b1, b2)
    assert int(b1.min()) >= 0 and int(b1.max()) < 64
    counts = torch.bincount(b1, minlength=64).float()
    # roughly uniform: no block more than 3x the mean
    assert float(counts.max()) < 3 * float(counts.mean())
