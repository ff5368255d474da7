"""Isolated A/B: K10 LDS histogram kernel vs torch scatter_add reference."""

import os
import sys
import time

import torch

sys.path.insert(0, ".")
from harmony_amd import ops  # noqa: E402


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def case(B, F, nb, n_nodes):
    g = torch.Generator().manual_seed(1)
    bins = torch.randint(0, nb, (B, F), generator=g).int().cuda()
    resid = torch.randn(B, generator=g).cuda()
    node = torch.randint(0, n_nodes, (B,), generator=g).int().cuda()
    t_hip = bench(lambda: ops.gbt_hist(bins, resid, node, n_nodes, nb))
    os.environ["HARMONY_FORCE_TORCH_OPS"] = "1"
    t_ref = bench(lambda: ops.gbt_hist(bins, resid, node, n_nodes, nb))
    del os.environ["HARMONY_FORCE_TORCH_OPS"]
    print(f"B={B:7d} F={F:3d} nb={nb} nodes={n_nodes:3d}: "
          f"hip {t_hip:7.3f} ms  torch {t_ref:7.3f} ms  "
          f"speedup {t_ref / t_hip:5.2f}x")


if __name__ == "__main__":
    case(4096, 32, 64, 8)        # app default, deepest level
    case(65536, 32, 64, 8)       # big batch
    case(65536, 128, 64, 8)      # wide features
    case(1048576, 32, 64, 8)     # 1M samples
    case(1048576, 32, 64, 128)   # deep level -> fc chunking
