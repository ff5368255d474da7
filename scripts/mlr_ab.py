"""Within-process A/B of MLR forward/grad variants (guide §5.4 rule 24:
interleaved rounds, report median)."""

import sys
import time

import torch

sys.path.insert(0, ".")
from harmony_amd import ops  # noqa: E402

hip = ops._load_hip()
assert hip is not None


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    ts.sort()
    return ts[len(ts) // 2] * 1e3  # median ms


def main():
    B, F, C = 16384, 16384, 10
    X = torch.randn(B, F, device="cuda")
    W = torch.randn(C, F, device="cuda") * 0.1
    y = torch.randint(0, C, (B,), device="cuda")
    P = torch.randn(B, C, device="cuda")
    Xt = X.t().contiguous()

    variants = {
        "fwd_gemm_BN": lambda: X @ W.t(),                       # M=B,N=C
        "fwd_gemm_CT": lambda: (W @ X.t()).t().contiguous(),    # M=C,N=B
        "fwd_fused_kernel": lambda: hip.mlr_fwd(X, W, y),
        "fwd_gemmCT_softmax": lambda: hip.mlr_softmax_grad(
            (W @ X.t()).t().contiguous(), y),
        "grad_gemm": lambda: P.t() @ X,
        "grad_kernel": lambda: hip.mlr_grad(P, X),
    }
    rounds = {k: [] for k in variants}
    for r in range(3):
        for k, fn in variants.items():
            rounds[k].append(bench(fn, iters=20, warmup=3))
    for k, ts in rounds.items():
        print(f"{k:22s} {min(ts):8.3f} ms (runs: {[f'{t:.3f}' for t in ts]})")


if __name__ == "__main__":
    main()
