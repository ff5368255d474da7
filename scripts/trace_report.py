"""Summarize tracer JSONL files (harmony_amd/utils/tracing.py output):
per-(job, phase) count/total/mean/p95 wall time — the quick look at where
a job's batches spend their time, like the reference's per-batch
pull/comp/push metric breakdown.

Usage: python scripts/trace_report.py <trace.jsonl> [more.jsonl ...]
"""

import json
import sys
from collections import defaultdict


def report(paths):
    acc = defaultdict(list)
    for p in paths:
        with open(p) as f:
            for line in f:
                try:
                    r = json.loads(line)
                except json.JSONDecodeError:
                    continue
                acc[(r.get("job", "?"), r["name"], r.get("rank", 0))].append(
                    r["dur_ms"])
    rows = []
    for (job, name, rank), ds in sorted(acc.items()):
        ds.sort()
        n = len(ds)
        rows.append((job, name, rank, n, sum(ds),
                     sum(ds) / n, ds[min(n - 1, int(round(0.95 * n)))]))
    w = max((len(f"{j}/{nm}") for j, nm, *_ in rows), default=10)
    print(f"{'job/span':<{w}}  rank    n   total ms    mean ms     p95 ms")
    for job, name, rank, n, tot, mean, p95 in rows:
        print(f"{job + '/' + name:<{w}}  {rank:4d} {n:4d} {tot:10.2f} "
              f"{mean:10.3f} {p95:10.3f}")


if __name__ == "__main__":
    if len(sys.argv) < 2:
        sys.exit(__doc__)
    report(sys.argv[1:])
