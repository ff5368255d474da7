#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db (rocpd schema) into per-kernel stats:
python scripts/rocpd_stats.py <results.db> [top_n]"""
import sqlite3
import sys


def kernel_stats(db, top=25):
    con = sqlite3.connect(db)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    uid = kd[len("rocpd_kernel_dispatch_"):]
    q = f"""
    SELECT ks.display_name AS name, COUNT(*) n,
           SUM(k.end - k.start) / 1e6 total_ms,
           AVG(k.end - k.start) / 1e3 avg_us
    FROM {kd} k
    JOIN rocpd_info_kernel_symbol_{uid} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY total_ms DESC LIMIT {top}
    """
    return list(con.execute(q))


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 25
    print(f"| total ms | calls | avg us | kernel |")
    print(f"|---:|---:|---:|---|")
    for name, n, tot, avg in kernel_stats(sys.argv[1], top):
        print(f"| {tot:.3f} | {n} | {avg:.1f} | `{name[:95]}` |")
