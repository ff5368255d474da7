"""CU-partitioned streams for multi-tenant GPU sharing.

MI355X has no MPS-style preemption: co-located jobs' full-chip kernels
timeshare all 256 CUs and evict each other's working sets from the
per-XCD L2s. `hipExtStreamCreateWithCUMask` pins a stream's kernels to a
CU subset — the MI355X-native analogue of the reference's per-executor
resource arbitration (LocalTaskUnitScheduler.java:33's CPU semaphores).
Partitions are allocated in contiguous CU runs so each job's CUs cluster
on as few XCDs as possible (private L2 locality).

Usage (bench/jobserver):
    streams = cu_partitioned_streams({"lda": 96, "nmf": 96, "mlr": 64})
    with torch.cuda.stream(streams["lda"]): ...
"""

from __future__ import annotations

from typing import Dict, List

import torch

_TOTAL_CUS = 256          # MI355X: 8 XCDs x 32 CUs
_created: List[int] = []  # raw handles (freed at process exit with the rest)


def cu_partitioned_streams(shares: Dict[str, int],
                           total_cus: int = _TOTAL_CUS
                           ) -> Dict[str, torch.cuda.ExternalStream]:
    """One CU-masked stream per entry; contiguous disjoint CU ranges in
    dict order. Shares are CU counts and must sum to <= total_cus."""
    from harmony_amd import ops

    hip = ops._load_hip()
    assert hip is not None and torch.cuda.is_available(), \
        "CU-masked streams need the HIP extension and a GPU"
    assert sum(shares.values()) <= total_cus, shares
    out = {}
    base = 0
    for name, n in shares.items():
        words = [0] * (total_cus // 32)
        for cu in range(base, base + n):
            words[cu // 32] |= (1 << (cu % 32))
        # torch int32 is signed: map the u32 bit patterns
        words = [w - (1 << 32) if w >= (1 << 31) else w for w in words]
        handle = hip.os_cu_masked_stream(
            torch.tensor(words, dtype=torch.int32))
        _created.append(handle)
        out[name] = torch.cuda.ExternalStream(handle)
        base += n
    return out
