"""2-process hipIpc prototype on one GPU: export, map, gather, scatter-add."""

import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, ".")


def child(q_h, q_done):
    try:
        _child(q_h, q_done)
    except Exception:  # noqa: BLE001
        import traceback

        q_done.put("CHILD FAILED:\n" + traceback.format_exc())


def _child(q_h, q_done):
    import torch

    from harmony_amd import ops

    hip = ops._load_hip()
    torch.cuda.set_device(0)
    handle = q_h.get(timeout=300)
    ptr = hip.os_ipc_open(torch.tensor(list(handle), dtype=torch.uint8))
    idx = torch.tensor([0, 2, 5], dtype=torch.int64, device="cuda")
    got = hip.os_gather(ptr, idx, 4, 0)
    torch.cuda.synchronize()
    assert abs(got[0, 0].item()) < 1e-6 and abs(got[1, 0].item() - 2) < 1e-6 \
        and abs(got[2, 3].item() - 5.3) < 1e-5, got.tolist()
    hip.os_scatter_add(ptr, idx, torch.full((3, 4), 10.0, device="cuda"))
    torch.cuda.synchronize()
    hip.os_ipc_close(ptr)
    q_done.put("child ok")


def main():
    import torch

    from harmony_amd import ops

    hip = ops._load_hip()
    torch.cuda.set_device(0)
    shard = hip.os_shard_alloc(8, 4, 0)
    pat = (torch.arange(8, device="cuda").float().unsqueeze(1)
           + torch.arange(4, device="cuda").float() * 0.1)
    shard.copy_(pat)
    torch.cuda.synchronize()
    h = hip.os_ipc_handle(shard)
    ctx = mp.get_context("spawn")
    q_h, q_done = ctx.Queue(), ctx.Queue()
    p = ctx.Process(target=child, args=(q_h, q_done))
    p.start()
    q_h.put(bytes(h.tolist()))
    print(q_done.get(timeout=300))
    p.join(timeout=30)
    torch.cuda.synchronize()
    # child's +10 scatter must be visible here
    assert abs(shard[2, 0].item() - 12.0) < 1e-6, shard[2, 0].item()
    assert abs(shard[5, 3].item() - 15.3) < 1e-6
    assert abs(shard[1, 0].item() - 1.0) < 1e-6   # untouched row
    print("parent ok: cross-process gather + scatter-add verified")


if __name__ == "__main__":
    main()
