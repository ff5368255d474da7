"""Isolated A/B: K11 persistent Lasso CD kernel vs the torch op-queue loop."""

import os
import sys
import time

import torch

sys.path.insert(0, ".")
from harmony_amd import ops  # noqa: E402


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def case(B, F):
    g = torch.Generator().manual_seed(1)
    X = torch.randn(B, F, generator=g).cuda()
    y = torch.randn(B, generator=g).cuda()
    w = torch.zeros(F, device="cuda")
    r = y.clone()
    col_sq = (X * X).sum(dim=0).clamp_min(1e-9)
    t_hip = bench(lambda: ops.lasso_cd(X, r, w, col_sq, 0.05 * B))
    os.environ["HARMONY_FORCE_TORCH_OPS"] = "1"
    t_ref = bench(lambda: ops.lasso_cd(X, r, w, col_sq, 0.05 * B),
                  iters=5)
    del os.environ["HARMONY_FORCE_TORCH_OPS"]
    print(f"B={B:6d} F={F:4d}: hip {t_hip:8.3f} ms  torch {t_ref:8.3f} ms  "
          f"speedup {t_ref / t_hip:6.1f}x")


if __name__ == "__main__":
    case(2048, 256)    # app default
    case(2048, 1024)
    case(8192, 256)
    case(14000, 512)   # near the LDS cap
