"""put/remove parity + cached model accessor (reference TableImpl multiPut/
remove, CachedModelAccessor)."""

import torch

from tests.dist_helper import run_dist


def test_put_remove_local():
    from harmony_amd.config import TableConfig
    from harmony_amd.et.table import Table

    cfg = TableConfig(table_id="pr", num_keys=32, value_dim=2, num_blocks=4,
                      init_fn="gaussian")
    t = Table(cfg, 0, 1, torch.device("cpu"))
    initial = t.shard.clone()
    keys = torch.tensor([1, 9, 30])
    t.put(keys, torch.ones(3, 2) * 7)
    assert torch.all(t.get(keys) == 7)
    t.remove(keys)
    # removed rows are back to the deterministic init values
    assert torch.equal(t.shard, initial)


def _put_remove_worker(rank, world):
    from harmony_amd.config import RuntimeConfig, TableConfig
    from harmony_amd.et.table import Table
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cfg = TableConfig(table_id="pr2", num_keys=32, value_dim=2, num_blocks=4,
                      init_fn="gaussian")
    t = Table(cfg, ctx.rank, ctx.world_size, ctx.device,
              comm=ctx.new_data_plane())
    if rank == 0:
        keys = torch.tensor([0, 20])
        vals = torch.full((2, 2), 5.0)
    else:
        keys = torch.tensor([3, 25])
        vals = torch.full((2, 2), 9.0)
    t.put(keys, vals)   # collective (both ranks put different keys)
    got = t.get(torch.tensor([0, 3, 20, 25]))
    ok = got[:, 0].tolist() == [5.0, 9.0, 5.0, 9.0]
    t.remove(torch.tensor([0, 3, 20, 25]))
    # back to deterministic init: both ranks agree on the value of key 0
    v0 = t.get(torch.tensor([0, 3, 20, 25]))
    return ok, v0.sum().item()


def test_put_remove_distributed():
    res = run_dist(_put_remove_worker, world=2)
    assert all(ok for ok, _ in res)
    assert abs(res[0][1] - res[1][1]) < 1e-6


def test_cached_model_accessor_refresh_and_writethrough():
    from harmony_amd.config import TableConfig
    from harmony_amd.dolphin.model_accessor import CachedModelAccessor
    from harmony_amd.et.table import Table

    cfg = TableConfig(table_id="cma", num_keys=8, value_dim=2, num_blocks=4)
    t = Table(cfg, 0, 1, torch.device("cpu"))
    acc = CachedModelAccessor(t, refresh_batches=2)
    v1 = acc.pull_all().clone()
    assert torch.all(v1 == 0)
    # external update not visible until refresh
    t.update(torch.tensor([0]), torch.ones(1, 2))
    v2 = acc.pull_all()
    assert torch.all(v2[0] == 0)        # cached
    v3 = acc.pull_all()                 # age hits refresh -> repull
    assert torch.all(v3[0] == 1)
    # push is applied write-through to the cache
    acc.push(torch.tensor([2]), torch.ones(1, 2) * 3)
    assert torch.all(acc.pull_all()[2] == 3)
    assert torch.all(t.get(torch.tensor([2])) == 3)
