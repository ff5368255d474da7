"""One-sided pull/push bandwidth microbench (2 procs sharing the GPU)."""

import datetime
import os
import sys
import time

sys.path.insert(0, ".")


def worker(rank, world):
    import torch

    from harmony_amd.config import TableConfig
    from harmony_amd.et.onesided import OneSidedTable
    from torch.distributed import TCPStore

    torch.cuda.set_device(0)
    store = TCPStore("127.0.0.1", int(os.environ["MASTER_PORT"]), world,
                     is_master=(rank == 0),
                     timeout=datetime.timedelta(seconds=120),
                     wait_for_workers=False)

    def barrier(name):
        store.add(f"b/{name}", 1)
        while int(store.get(f"b/{name}")) < world:
            time.sleep(0.002)

    N, K = 1 << 20, 64                       # 1M keys x 64 f32 = 256 MB
    cfg = TableConfig(table_id="bw", num_keys=N, value_dim=K,
                      num_blocks=64, update_fn="add", init_fn="zeros")
    t = OneSidedTable(cfg, rank, world, torch.device("cuda"), store=store)
    barrier("alloc")
    t.connect()
    barrier("conn")
    keys = torch.randint(0, N, (1 << 18,), device="cuda")  # 256k random rows
    d = torch.ones(keys.shape[0], K, device="cuda")
    mb = keys.shape[0] * K * 4 / 1e6
    for name, fn in (("pull", lambda: t.pull(keys)),
                     ("push", lambda: t.push(keys, d))):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        if rank == 0:
            print(f"{name}: {mb:.0f} MB in {dt*1e3:.2f} ms = "
                  f"{mb/1e3/dt:.1f} GB/s (random rows, half remote)")
    barrier("done")
    t.close()


if __name__ == "__main__":
    import multiprocessing as mp

    os.environ.setdefault("MASTER_PORT", "29877")
    ctx = mp.get_context("spawn")
    ps = [ctx.Process(target=worker, args=(r, 2)) for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(timeout=300)
