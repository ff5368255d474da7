"""Collective-order sanitizer.

The reference relies on safety-by-construction (@GuardedBy annotations,
per-block comm-thread serialization) and has no runtime sanitizer; the
survey explicitly recommends the rebuild add one. Our analogue of its
data-race class is COLLECTIVE-ORDER divergence: if two ranks that co-run
jobs enqueue those jobs' NET phases (RCCL collectives) in different
relative orders, RCCL deadlocks or silently corrupts buffers.

Enable with HARMONY_SANITIZE=1: every NET phase a rank enters is appended
to a per-rank log in the control store ("san/<rank>" = "job@phase#ticket;"
records). `validate(store, world_size)` then checks the two invariants the
ticket protocol must guarantee:

  1. per-rank ticket monotonicity — each rank issues NET phases in strictly
     increasing global-ticket order (filtered to its jobs);
  2. cross-rank agreement — every (job, phase) that appears on more than
     one rank drew the SAME global ticket everywhere.

Together these imply: any two ranks order their shared jobs' phases
identically, which is exactly the RCCL deadlock-freedom condition.
"""

from __future__ import annotations

import os
from typing import Dict, List, Tuple


def enabled() -> bool:
    return os.environ.get("HARMONY_SANITIZE") == "1"


def record(store, rank: int, job_id: str, phase_idx: int, seq: int) -> None:
    """Append one NET-phase entry to this rank's sanitizer log."""
    store.append(f"san/{rank}", f"{job_id}@{phase_idx}#{seq};")


def _parse(raw: str) -> List[Tuple[str, int, int]]:
    out = []
    for rec in raw.split(";"):
        if not rec:
            continue
        jp, seq = rec.rsplit("#", 1)
        job, phase = jp.rsplit("@", 1)
        out.append((job, int(phase), int(seq)))
    return out


def validate(store, world_size: int) -> List[str]:
    """Returns a list of violation messages (empty = clean)."""
    logs: Dict[int, List[Tuple[str, int, int]]] = {}
    for r in range(world_size):
        try:
            raw = store.get(f"san/{r}").decode()
        except Exception:  # rank recorded nothing
            raw = ""
        logs[r] = _parse(raw)
    errs: List[str] = []
    ticket_of: Dict[Tuple[str, int], Tuple[int, int]] = {}
    for r, entries in logs.items():
        last = -1
        for job, phase, seq in entries:
            if seq <= last:
                errs.append(f"rank {r}: ticket order violated — "
                            f"{job}@{phase} drew #{seq} after #{last}")
            last = seq
            key = (job, phase)
            if key in ticket_of:
                r0, s0 = ticket_of[key]
                if s0 != seq:
                    errs.append(f"{job}@{phase}: rank {r0} drew #{s0} but "
                                f"rank {r} drew #{seq} — ranks would enqueue "
                                f"collectives in different orders")
            else:
                ticket_of[key] = (r, seq)
    return errs
