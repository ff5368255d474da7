"""Multi-rank (gloo, CPU) tests of the collective data plane — the same code
path RCCL takes on the GPU box (backend-adaptive inside DataPlane)."""

import torch

from tests.dist_helper import run_dist


def _mk(rank, world, **kw):
    from harmony_amd.config import RuntimeConfig, TableConfig
    from harmony_amd.et.table import Table
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    cfg = TableConfig(table_id=kw.pop("table_id", "t"),
                      num_keys=kw.pop("num_keys", 64),
                      value_dim=kw.pop("value_dim", 4),
                      num_blocks=kw.pop("num_blocks", 8), **kw)
    comm = ctx.new_data_plane()
    table = Table(cfg, ctx.rank, ctx.world_size, ctx.device, comm=comm)
    return ctx, table


def _pull_all_worker(rank, world):
    ctx, table = _mk(rank, world)
    # each rank writes rank+1 into its own rows, then pull_all
    table.shard.fill_(float(rank + 1))
    full = table.pull_all()
    # rows of rank 0's blocks are 1.0, rank 1's are 2.0
    owned0 = table.ownership.owned_blocks(0)
    bs = table.block_size
    ok = all(bool((full[b * bs:(b + 1) * bs] == (table.ownership.owner_of_int(b) + 1)).all())
             for b in range(table.cfg.num_blocks))
    return ok


def test_pull_all():
    assert all(run_dist(_pull_all_worker, world=2))


def _pull_keys_worker(rank, world):
    ctx, table = _mk(rank, world)
    # fill each owned row with its global key value
    for b in table.owned_blocks:
        view = table.local_block_view(b)
        keys = torch.arange(b * table.block_size, (b + 1) * table.block_size)
        view.copy_(keys.float().unsqueeze(1).repeat(1, 4))
    keys = torch.tensor([1, 8, 17, 33, 63, 5]) if rank == 0 else \
        torch.tensor([62, 0, 31, 32])
    vals = table.get(keys)
    return bool(torch.allclose(vals, keys.float().unsqueeze(1).repeat(1, 4)))


def test_pull_keys_cross_rank():
    assert all(run_dist(_pull_keys_worker, world=2))


def _push_keys_worker(rank, world):
    ctx, table = _mk(rank, world)
    # both ranks push +1 to the same keys (and rank-specific keys)
    keys = torch.tensor([3, 40, 3])          # dup key 3 within a rank too
    deltas = torch.ones(3, 4)
    table.update(keys, deltas)
    full = table.pull_all()
    # key 3: 2 ranks x 2 dups = +4; key 40: 2 ranks x 1 = +2
    return (float(full[3, 0]), float(full[40, 0]), float(full[10, 0]))


def test_push_keys_aggregates_across_ranks():
    res = run_dist(_push_keys_worker, world=2)
    for k3, k40, k10 in res:
        assert k3 == 4.0
        assert k40 == 2.0
        assert k10 == 0.0


def _push_dense_worker(rank, world):
    ctx, table = _mk(rank, world)
    grad = torch.full((table.cfg.padded_num_keys, 4), float(rank + 1))
    table.push_dense(grad)   # sum = 3.0 everywhere
    full = table.pull_all()
    return bool((full == 3.0).all())


def test_push_dense_reduces():
    assert all(run_dist(_push_dense_worker, world=2))


def _mlr_job_worker(rank, world):
    from harmony_amd.config import JobConfig, RuntimeConfig
    from harmony_amd.dolphin.master import run_job
    from harmony_amd.runtime.bootstrap import init_executor

    ctx = init_executor(RuntimeConfig(device="cpu"))
    job = JobConfig(job_id="j2_mlr", app="mlr", max_num_epochs=2,
                    num_mini_batches=2,
                    app_args={"num_classes": 4, "num_features": 32,
                              "num_parts_per_class": 2, "batch_size": 128,
                              "step_size": 0.5})
    m = run_job(job, ctx)
    s = m.summary()
    return (s["num_batches"], s["accuracy"])


def test_mlr_two_ranks():
    res = run_dist(_mlr_job_worker, world=2)
    for nb, acc in res:
        assert nb == 4
        assert acc > 0.4


def _edge_cases_worker(rank, world):
    # degenerate shapes the all-to-all-v plumbing must survive:
    # empty key sets on one rank, single-key, all-keys-one-owner,
    # duplicated keys through the merge path, then remove/put round-trips
    ctx, table = _mk(rank, world)
    out = []

    # 1. one rank pulls nothing while the other pulls (collective pair)
    keys = (torch.tensor([], dtype=torch.int64) if rank == 0
            else torch.tensor([3]))
    vals = table.get(keys)
    out.append(vals.shape[0] == keys.shape[0])

    # 2. all keys owned by ONE rank (0 owns blocks 0-3 = keys 0-31)
    keys = torch.tensor([0, 1, 2, 30, 31])
    vals = table.get(keys)
    out.append(vals.shape == (5, 4))

    # 3. duplicated keys in an update: merge path sums per key first
    keys = torch.tensor([10, 10, 40, 10])
    deltas = torch.ones(4, 4)
    table.update(keys, deltas)
    got = table.get(torch.tensor([10, 40]))
    # both ranks pushed: key 10 got 3 per rank, key 40 got 1 per rank
    out.append(bool(torch.allclose(
        got, torch.tensor([[6.0], [2.0]]).repeat(1, 4))))

    # 4. put then remove restores deterministic init (zeros)
    table.put(torch.tensor([7]), torch.full((1, 4), 9.0))
    out.append(float(table.get(torch.tensor([7]))[0, 0]) == 9.0)
    table.remove(torch.tensor([7]))
    out.append(float(table.get(torch.tensor([7]))[0, 0]) == 0.0)
    return out


def test_comm_edge_cases_two_ranks():
    res = run_dist(_edge_cases_worker, world=2)
    for r in res:
        assert all(r), res


def _global_empty_worker(rank, world):
    # globally-empty exchanges (every rank empty) must be symmetric no-ops
    # on every path — an asymmetric skip would strand peers in a collective
    ctx, table = _mk(rank, world)
    import torch as T

    e = T.tensor([], dtype=T.int64)
    out = table.get(e)
    ok = [out.shape == (0, 4)]
    table.update(e, T.empty(0, 4))
    table.put(e, T.empty(0, 4))
    table.remove(e)
    # and a mixed call right after (plumbing still healthy)
    v = table.get(T.tensor([1]))
    ok.append(v.shape == (1, 4))
    return ok


def test_global_empty_exchanges():
    res = run_dist(_global_empty_worker, world=2)
    for r in res:
        assert all(r), res
