"""RCCL-path smoke at world 1 on a real GPU: the `backend == "nccl"`
branches of the data plane (all_gather_into_tensor, reduce_scatter_tensor,
all_to_all_single with device split sizes, all_gather count exchange)
never run under the CPU/gloo tests — this executes those exact RCCL API
calls with device tensors so argument/contiguity/dtype bugs can't wait
for the first 8-GPU run to surface. Reference behavior:
TableImpl multiGet/multiUpdate (et/evaluator/impl/TableImpl.java:284,460).
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_dataplane_nccl_world1():
    import torch.distributed as dist

    from harmony_amd.config import TableConfig
    from harmony_amd.et.comm import DataPlane
    from harmony_amd.et.table import Table

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        dev = torch.device("cuda:0")
        torch.cuda.set_device(dev)
        cfg = TableConfig(table_id="nccl1", num_keys=64, value_dim=8,
                          num_blocks=8, update_fn="add", init_fn="zeros")
        t = Table(cfg, 0, 1, dev)
        dp = DataPlane(None, 0, 1, dev)
        assert dp.backend == "nccl"
        # pull_all -> all_gather_into_tensor path
        full = dp.pull_all(t)
        assert full.shape[0] >= 64 and bool((full == 0).all())
        # push_dense -> reduce_scatter_tensor path
        g = torch.ones(full.shape, device=dev)
        dp.push_dense(t, g)
        assert bool((dp.pull_all(t)[:64] == 1).all())
        # sparse pull/push -> count all_gather + all_to_all_single paths
        keys = torch.arange(0, 64, 3, device=dev)
        vals = dp.pull_keys(t, keys)
        assert bool((vals == 1).all())
        dp.push_keys(t, keys, torch.full((keys.numel(), 8), 2.0, device=dev))
        assert bool((dp.pull_keys(t, keys) == 3).all())
        # piggyback variant (Pregel halt vote fuses into this exchange)
        pig = dp.push_keys(t, keys,
                           torch.full((keys.numel(), 8), 1.0, device=dev),
                           piggyback=torch.tensor([5, 7]))
        assert pig is not None and pig.tolist() == [5, 7]
    finally:
        dist.destroy_process_group()
