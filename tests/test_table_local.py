import torch

from harmony_amd.config import TableConfig
from harmony_amd.et.table import ObjectTable, Table


def make_table(**kw):
    cfg = TableConfig(table_id=kw.pop("table_id", "t"), num_keys=kw.pop("num_keys", 64),
                      value_dim=kw.pop("value_dim", 4), num_blocks=kw.pop("num_blocks", 8),
                      **kw)
    return Table(cfg, rank=0, world_size=1, device=torch.device("cpu"))


def test_get_put_update_roundtrip():
    t = make_table()
    keys = torch.tensor([0, 5, 13, 63])
    assert torch.all(t.get(keys) == 0)
    vals = torch.randn(4, 4)
    t.put_local(keys, vals)
    assert torch.allclose(t.get(keys), vals)
    t.update(keys, torch.ones(4, 4))
    assert torch.allclose(t.get(keys), vals + 1)


def test_update_aggregates_duplicate_keys():
    t = make_table()
    keys = torch.tensor([3, 3, 3, 7])
    deltas = torch.ones(4, 4)
    t.update(keys, deltas)
    assert torch.allclose(t.get(torch.tensor([3])), torch.full((1, 4), 3.0))
    assert torch.allclose(t.get(torch.tensor([7])), torch.ones(1, 4))


def test_nmf_update_fn_clamps():
    t = make_table(update_fn="nmf_sgd", init_fn="uniform_clamped",
                   update_args={"step_size": 1.0})
    keys = torch.tensor([0, 1])
    before = t.get(keys).clone()
    big = torch.full((2, 4), 1e9)
    t.update(keys, big)  # new = clamp(old - 1.0*1e9, 0, max) == 0
    assert torch.all(t.get(keys) == 0)
    assert torch.all(before >= 0)


def test_deterministic_init_per_block():
    t1 = make_table(table_id="m", init_fn="gaussian")
    t2 = make_table(table_id="m", init_fn="gaussian")
    assert torch.equal(t1.shard, t2.shard)
    # different table id -> different init
    t3 = make_table(table_id="other", init_fn="gaussian")
    assert not torch.equal(t1.shard, t3.shard)


def test_block_views_and_migration_rebuild():
    t = make_table()
    keys = torch.arange(64)
    t.put_local(keys, torch.arange(64).float().unsqueeze(1).repeat(1, 4))
    blk2 = t.local_block_view(2).clone()
    dropped = t.drop_blocks([2])
    assert torch.equal(dropped[2], blk2)
    assert 2 not in t.owned_blocks
    t.adopt_blocks({2: blk2})
    assert torch.equal(t.local_block_view(2), blk2)
    # all data intact after drop+adopt
    assert torch.allclose(t.get(keys),
                          torch.arange(64).float().unsqueeze(1).repeat(1, 4))


def test_object_table():
    from harmony_amd.config import TableConfig

    cfg = TableConfig(table_id="gbt", num_keys=4, num_blocks=4, storage="object")
    t = ObjectTable(cfg, rank=0, world_size=1,
                    init_value=lambda k: [],
                    update_value=lambda v, d: v + [d])
    t.update(1, "tree_a")
    t.update(1, "tree_b")
    assert t.get(1) == ["tree_a", "tree_b"]
    assert t.get(2) == []
