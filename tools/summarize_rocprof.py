#!/usr/bin/env python3
"""Summarize rocprofv3 output (kernel stats CSVs and PMC counter CSVs) into a
compact per-kernel table for profiles/.

Usage: python tools/summarize_rocprof.py <dir-or-csv> [...]

Handles:
  *_kernel_stats.csv / *stats*.csv  (from --kernel-trace --stats)
  *_counter_collection.csv          (from --pmc X): aggregates Counter_Value
                                    per (Kernel_Name, Counter_Name), with
                                    per-dispatch mean
"""
import csv
import glob
import os
import sys
from collections import defaultdict


def short(name, n=70):
    name = name.split("(")[0]
    return name if len(name) <= n else name[:n] + "…"


def do_stats(path):
    with open(path) as fh:
        rows = list(csv.DictReader(fh))
    if not rows:
        return
    print(f"\n== {path} ==")
    cols = rows[0].keys()
    namec = next((c for c in cols if "Name" in c), None)
    for r in rows:
        print("  ".join(f"{k}={short(str(r[k]),60)}" for k in cols if r.get(k))
              if namec is None else
              f"{short(r[namec])}: " + "  ".join(
                  f"{k}={r[k]}" for k in cols if k != namec))


def do_counters(path):
    agg = defaultdict(lambda: [0.0, 0])
    with open(path) as fh:
        for r in csv.DictReader(fh):
            name = r.get("Kernel_Name") or r.get("Kernel-Name") or "?"
            cname = r.get("Counter_Name") or r.get("Counter-Name") or "?"
            try:
                val = float(r.get("Counter_Value") or r.get("Counter-Value"))
            except (TypeError, ValueError):
                continue
            key = (short(name), cname)
            agg[key][0] += val
            agg[key][1] += 1
    if not agg:
        return
    print(f"\n== {path} ==")
    for (kname, cname), (tot, n) in sorted(agg.items()):
        print(f"{kname} :: {cname}: total={tot:.6g} dispatches={n} "
              f"mean/dispatch={tot / n:.6g}")


def main(paths):
    files = []
    for p in paths:
        if os.path.isdir(p):
            files += glob.glob(os.path.join(p, "**", "*.csv"), recursive=True)
        else:
            files.append(p)
    for fp in sorted(files):
        base = os.path.basename(fp)
        if "counter" in base:
            do_counters(fp)
        else:
            do_stats(fp)


if __name__ == "__main__":
    main(sys.argv[1:])
