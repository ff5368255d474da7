"""Per-job logging (reference jobserver JobLogger: a per-job prefix so one
server's interleaved logs are attributable)."""

from __future__ import annotations

import logging
import sys


def job_logger(job_id: str, rank: int = 0) -> logging.Logger:
    """Logger named for the job, emitting '[job_id rN] msg' lines."""
    name = f"harmony.job.{job_id}"
    lg = logging.getLogger(name)
    if not lg.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(
            f"[{job_id} r{rank}] %(levelname)s %(message)s"))
        lg.addHandler(h)
        lg.setLevel(logging.INFO)
        lg.propagate = False
    return lg
