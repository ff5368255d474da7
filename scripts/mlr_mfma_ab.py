#!/usr/bin/env python3
"""A/B: MLR step — rocBLAS GEMM pair vs the fused f32-MFMA kernels.

Bench shape (BASELINE config): B=16384, F=16384, C=10, fp32. The step is
X-bandwidth-bound (X = 1 GiB, read by fwd AND grad). Variants:
  rocblas      X@W^T (Tensile) + fused softmax kernel + P^T@X (Tensile)
  mfma_rb0     fused MFMA pair, whole batch per launch (X from HBM twice)
  mfma_rbN     row-blocked: fwd+grad per N-row block -> grad re-reads X
               from the 256 MiB Infinity Cache
Numerics: MFMA f32 is an exact fmaf chain; compare grads vs the rocBLAS
path (summation order differs -> rtol).

Run on a GPU box:  python scripts/mlr_mfma_ab.py [--iters 30]
"""
import argparse
import time

import torch

from harmony_amd import ops


def bench(fn, iters, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--B", type=int, default=16384)
    ap.add_argument("--F", type=int, default=16384)
    ap.add_argument("--C", type=int, default=10)
    ap.add_argument("--sweep", action="store_true")
    args = ap.parse_args()
    if args.sweep:
        sweep()
        return
    torch.manual_seed(0)
    dev = "cuda"
    X = torch.randn(args.B, args.F, device=dev)
    W = torch.randn(args.C, args.F, device=dev) * 0.01
    y = torch.randint(0, args.C, (args.B,), device=dev)

    def roc():
        p, loss, corr = ops.mlr_forward(X, W, y)
        return ops.mlr_grad_gemm(p, X), loss, corr

    g0, l0, c0 = roc()
    out = {"rocblas_pair_ms": bench(roc, args.iters)}
    wt = torch.zeros((args.F, 16), device=dev)
    for rb in (0, 1024, 2048, 4096):
        if rb and args.B % rb:
            continue
        g1, l1, c1 = ops.mlr_step_mfma(X, W, y, row_block=rb, Wt_buf=wt)
        rel = ((g1 - g0).abs().max() / g0.abs().max()).item()
        dl = abs(l1.item() - l0.item()) / max(1.0, abs(l0.item()))
        assert rel < 1e-3, f"grad mismatch rb={rb}: rel={rel}"
        assert dl < 1e-4, f"loss mismatch rb={rb}: {l1.item()} vs {l0.item()}"
        assert int(c1) == int(c0), (int(c1), int(c0))
        ms = bench(lambda rb=rb: ops.mlr_step_mfma(X, W, y, row_block=rb,
                                                   Wt_buf=wt), args.iters)
        out[f"mfma_rb{rb}_ms"] = ms
        out[f"mfma_rb{rb}_grad_rel"] = rel
    for rb in (1024, 2048, 4096):
        if args.B % rb:
            continue
        g1, l1, c1 = ops.mlr_step_mfma_pipelined(X, W, y, row_block=rb,
                                                 Wt_buf=wt)
        rel = ((g1 - g0).abs().max() / g0.abs().max()).item()
        assert rel < 1e-3, f"pipe grad mismatch rb={rb}: {rel}"
        assert int(c1) == int(c0)
        out[f"mfma_pipe_rb{rb}_ms"] = bench(
            lambda rb=rb: ops.mlr_step_mfma_pipelined(X, W, y, row_block=rb,
                                                      Wt_buf=wt), args.iters)
    xbytes = args.B * args.F * 4
    out["hbm_floor_ms_2pass"] = 2 * xbytes / 6.3e12 * 1e3
    out["hbm_floor_ms_1pass"] = xbytes / 6.3e12 * 1e3
    for k, v in out.items():
        print(f"{k:26s} {v:.4f}" if isinstance(v, float) else f"{k}: {v}")


def sweep():
    """Split/occupancy sweep + per-kernel timing (run with --sweep)."""
    import itertools
    torch.manual_seed(0)
    dev = "cuda"
    B = F = 16384
    X = torch.randn(B, F, device=dev)
    W = torch.randn(10, F, device=dev) * 0.01
    y = torch.randint(0, 10, (B,), device=dev)
    wt = torch.zeros((F, 16), device=dev)
    for rb in (0, 1024, 2048):
        for sf, sb in itertools.product((1, 2, 4, 8, 16, 32),
                                        (2, 4, 8, 16, 32)):
            if rb == 0 and (sf > 8 or sb > 16):
                continue
            if rb and rb // 64 < sb:
                continue
            try:
                ms = bench(lambda: ops.mlr_step_mfma(
                    X, W, y, row_block=rb, splitf=sf, splitb=sb, Wt_buf=wt),
                    12)
                print(f"rb={rb:5d} splitf={sf} splitb={sb:2d}  {ms:.4f} ms")
            except Exception as e:
                print(f"rb={rb} sf={sf} sb={sb}: {str(e)[:60]}")


if __name__ == "__main__":
    main()
