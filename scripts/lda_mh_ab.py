"""Isolated A/B of the LDA MH sweep kernel (bench-shaped data)."""

import os
import sys
import time

import torch

sys.path.insert(0, ".")
from harmony_amd import ops  # noqa: E402

hip = ops._load_hip()


def main():
    D, T, K, V = 16384, 128, 256, 100000
    g = torch.Generator().manual_seed(1)
    u = torch.rand(D * T, generator=g)
    w = (u * u * V).long().clamp_(0, V - 1).view(D, T).sort(1).values.reshape(-1)
    uw, wl = torch.unique(w, return_inverse=True)
    R = uw.shape[0]
    wt = torch.zeros(R, K, dtype=torch.int32)
    z = torch.randint(0, K, (D * T,), generator=g, dtype=torch.int32)
    wt.view(-1).scatter_add_(0, wl * K + z.long(),
                             torch.ones(D * T, dtype=torch.int32))
    nk = torch.bincount(z.long(), minlength=K).int()
    dt = torch.zeros(D, K, dtype=torch.int32)
    tok_doc = torch.arange(D).repeat_interleave(T)
    dt.view(-1).scatter_add_(0, tok_doc * K + z.long(),
                             torch.ones(D * T, dtype=torch.int32))
    dev = "cuda"
    wt, nk, dt = wt.to(dev), nk.to(dev), dt.to(dev)
    wl64 = wl.to(dev)
    offs = torch.arange(0, (D + 1) * T, T, dtype=torch.int64, device=dev)
    z = z.to(dev)
    import os as _os

    for bw in ("1", "2", "4", "8"):
        _os.environ["HARMONY_LDA_BUILD_WAVES"] = bw
        for _ in range(2):
            hip.lda_alias_build(wt, nk, 0.01, V)
        torch.cuda.synchronize()
        ts = []
        for _ in range(10):
            t0 = time.perf_counter()
            hip.lda_alias_build(wt, nk, 0.01, V)
            torch.cuda.synchronize()
            ts.append(time.perf_counter() - t0)
        ts.sort()
        print(f"alias_build waves={bw}: {ts[len(ts)//2]*1e3:.3f} ms")
    _os.environ.pop("HARMONY_LDA_BUILD_WAVES", None)
    tabs = hip.lda_alias_build(wt, nk, 0.01, V)
    prob, alias, tp, ta, qv, qsum, invden = tabs
    torch.cuda.synchronize()
    print(f"uniq words {R}")
    for thr, pf in (("128", "0"), ("128", "1"), ("64", "0"), ("64", "1"),
                    ("256", "1")):
        os.environ["HARMONY_LDA_MH_THREADS"] = thr
        os.environ["HARMONY_LDA_MH_PREFETCH"] = pf
        zz = z.clone()
        dtt = dt.clone()
        for _ in range(3):
            hip.lda_mh(dtt, wt, invden, prob, alias, tp, ta, qv, offs,
                       wl64, zz, 0.1, 0.01, 42)
        torch.cuda.synchronize()
        ts = []
        for i in range(15):
            t0 = time.perf_counter()
            hip.lda_mh(dtt, wt, invden, prob, alias, tp, ta, qv, offs,
                       wl64, zz, 0.1, 0.01, 43 + i)
            torch.cuda.synchronize()
            ts.append(time.perf_counter() - t0)
        ts.sort()
        print(f"threads={thr} prefetch={pf}: {ts[len(ts)//2]*1e3:.3f} ms")


if __name__ == "__main__":
    main()
