"""Isolated A/B of NMF gradient variants + LDA sampler timing on MI355X."""

import sys
import time

import torch

sys.path.insert(0, ".")
from harmony_amd import ops  # noqa: E402

hip = ops._load_hip()
assert hip is not None


def bench(fn, iters=20, warmup=4):
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
    return ts[len(ts) // 2] * 1e3


def main():
    torch.manual_seed(0)
    # bench default shapes: 16384 rows x 128 nnz, rank 100, 65536 cols
    n, k, nnz_per_row, m = 16384, 100, 128, 65536
    nnz = n * nnz_per_row
    L = torch.rand(n, k, device="cuda")
    row_ptr = torch.arange(0, nnz + 1, nnz_per_row, device="cuda")
    col_g = torch.randint(0, m, (nnz,), device="cuda")
    uniq, col = torch.unique(col_g, return_inverse=True)
    R = torch.rand(uniq.shape[0], k, device="cuda")
    vals = torch.rand(nnz, device="cuda")
    perm = torch.argsort(col, stable=True)
    counts = torch.bincount(col, minlength=uniq.shape[0])
    seg_ptr = torch.zeros(uniq.shape[0] + 1, dtype=torch.int64, device="cuda")
    seg_ptr[1:] = counts.cumsum(0)
    row_of = torch.repeat_interleave(torch.arange(n, device="cuda"),
                                     row_ptr[1:] - row_ptr[:-1])
    row_sorted = row_of[perm]

    print("nnz:", nnz, "uniq cols:", uniq.shape[0])
    variants = {
        "nmf_onepass_atomic": lambda: hip.nmf_grad(L, R, row_ptr, col, vals, 0.0),
        "nmf_twopass": lambda: hip.nmf_grad_twopass(
            L, R, row_ptr, col, vals, perm, seg_ptr, row_sorted, 0.0),
    }
    # LDA: bench default 8192 docs x 128 tokens, K=256, V=100k, Zipf sorted
    D, T, K, V = 8192, 128, 256, 100000
    g = torch.Generator().manual_seed(1)
    u = torch.rand(D * T, generator=g)
    w = (u * u * V).long().clamp_(0, V - 1).view(D, T).sort(dim=1).values.reshape(-1)
    uw, wl = torch.unique(w, return_inverse=True)
    wl = wl.cuda()
    z = torch.randint(0, K, (D * T,), dtype=torch.int32).cuda()
    offs = torch.arange(0, (D + 1) * T, T, device="cuda")
    dt = torch.zeros(D, K, dtype=torch.int32, device="cuda")
    dt.view(-1).scatter_add_(0, (torch.arange(D, device="cuda")
                                 .repeat_interleave(T) * K + z.long()),
                             torch.ones(D * T, dtype=torch.int32, device="cuda"))
    wt = torch.zeros(uw.shape[0], K, dtype=torch.int32, device="cuda")
    wt.view(-1).scatter_add_(0, wl * K + z.long(),
                             torch.ones(D * T, dtype=torch.int32, device="cuda"))
    ts_sum = wt.sum(0).to(torch.int32)
    variants["lda_gibbs_sorted"] = lambda: hip.lda_gibbs(
        dt, wt, ts_sum, offs, wl, z, 0.1, 0.01, V, 42)
    # unsorted tokens for comparison (cache-locality delta)
    w2 = (u * u * V).long().clamp_(0, V - 1)
    uw2, wl2 = torch.unique(w2, return_inverse=True)
    wl2 = wl2.cuda()
    z2 = z.clone()
    wt2 = torch.zeros(uw2.shape[0], K, dtype=torch.int32, device="cuda")
    wt2.view(-1).scatter_add_(0, wl2 * K + z2.long(),
                              torch.ones(D * T, dtype=torch.int32, device="cuda"))
    variants["lda_gibbs_unsorted"] = lambda: hip.lda_gibbs(
        dt, wt2, ts_sum, offs, wl2, z2, 0.1, 0.01, V, 42)

    rounds = {kk: [] for kk in variants}
    for r in range(3):
        for kk, fn in variants.items():
            rounds[kk].append(bench(fn))
    for kk, tt in rounds.items():
        print(f"{kk:22s} {min(tt):8.3f} ms (runs: {[f'{t:.3f}' for t in tt]})")


if __name__ == "__main__":
    main()
