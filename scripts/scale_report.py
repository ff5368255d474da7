"""Analyze multi-GPU scaling results (the driver's SCALE_rNN.json or any
collection of bench.py JSON lines at different N).

Accepts: a single JSON file containing a list of bench dicts, a file of
JSON lines, or several files each holding one dict. Prints per-N
throughput, weak-scaling efficiency vs N=1, and the per-step time deltas
that localize where scaling loss comes from (fixed control overhead vs
growing per-rank comm).

Usage: python scripts/scale_report.py SCALE_r01.json [more...]
"""

import json
import sys


def load(paths):
    recs = []
    for p in paths:
        with open(p) as f:
            txt = f.read().strip()
        try:
            obj = json.loads(txt)
            recs.extend(obj if isinstance(obj, list) else [obj])
        except json.JSONDecodeError:
            for line in txt.splitlines():
                line = line.strip()
                if line.startswith("{"):
                    try:
                        recs.append(json.loads(line))
                    except json.JSONDecodeError:
                        pass
    out = {}
    for r in recs:
        if "n_gpus" in r and "value" in r:
            out[int(r["n_gpus"])] = r
    return dict(sorted(out.items()))


def main(paths):
    by_n = load(paths)
    if not by_n:
        sys.exit("no bench records found")
    base = by_n.get(1)
    print(f"{'N':>3} {'Mex/s':>10} {'ms/step':>9} {'weak-eff':>9} "
          f"{'dt vs N=1 (ms)':>15}")
    for n, r in by_n.items():
        v, ms = r["value"], r["ms_per_step"]
        if base:
            eff = v / (base["value"] * n)
            dt = ms - base["ms_per_step"]
            print(f"{n:>3} {v / 1e6:>10.1f} {ms:>9.3f} {eff:>8.1%} "
                  f"{dt:>+15.3f}")
        else:
            print(f"{n:>3} {v / 1e6:>10.1f} {ms:>9.3f} {'n/a':>9} {'n/a':>15}")
    if base and len(by_n) > 2:
        # constant dt across N -> fixed per-step overhead (control plane);
        # dt growing with N -> per-rank comm cost (ring/all-to-all terms)
        ns = [n for n in by_n if n > 1]
        dts = [by_n[n]["ms_per_step"] - base["ms_per_step"] for n in ns]
        if max(dts) - min(dts) < 0.2 * max(abs(d) for d in dts + [1e-9]):
            print("\ndt roughly CONSTANT across N: fixed per-step overhead "
                  "(suspect the ticket control plane — see ROADMAP round-2 "
                  "plan item 3).")
        else:
            print("\ndt GROWS with N: per-rank communication term "
                  "(suspect collective sizes/topology — ROADMAP item 1).")


if __name__ == "__main__":
    if len(sys.argv) < 2:
        sys.exit(__doc__)
    main(sys.argv[1:])
