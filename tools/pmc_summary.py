#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc CSV output into a per-kernel counter table.

Usage: pmc_summary.py <dir-with-*_counter_collection.csv>
Sums each counter per kernel (demangled, truncated) and prints a table
plus derived HBM bytes (FETCH_SIZE+WRITE_SIZE are in 32B/KiB units per
the guide: hbm_bytes = (FETCH_SIZE + WRITE_SIZE) * 1024 when both
collected; we report the raw sums and the per-kernel share)."""
import csv
import glob
import re
import sys
from collections import defaultdict

d = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/pmc"
files = glob.glob(f"{d}/**/*counter_collection.csv", recursive=True)
if not files:
    sys.exit(f"no counter_collection.csv under {d}")

agg = defaultdict(lambda: defaultdict(float))   # kernel -> counter -> sum
calls = defaultdict(set)
for path in files:
    with open(path) as fh:
        r = csv.DictReader(fh)
        cols = {c.lower(): c for c in r.fieldnames}
        kn = cols.get("kernel_name")
        cn = cols.get("counter_name")
        cv = cols.get("counter_value")
        did = cols.get("dispatch_id") or cols.get("correlation_id")
        if not (kn and cn and cv):
            sys.exit(f"unexpected columns in {path}: {r.fieldnames}")
        for row in r:
            name = re.sub(r"<[^>]*>", "", row[kn]).split("(")[0]
            name = name.replace("void ", "").strip()
            agg[name][row[cn]] += float(row[cv])
            if did:
                calls[name].add(row[did])

counters = sorted({c for v in agg.values() for c in v})
order = sorted(agg, key=lambda k: -agg[k].get("SQ_WAVE_CYCLES", 0))
print(f"{'kernel':<28} {'calls':>7} " +
      " ".join(f"{c.replace('SQ_', ''):>16}" for c in counters))
for k in order:
    print(f"{k[:28]:<28} {len(calls[k]):>7} " +
          " ".join(f"{agg[k].get(c, 0):>16.3e}" for c in counters))

tot = {c: sum(v.get(c, 0) for v in agg.values()) for c in counters}
print(f"{'TOTAL':<28} {'':>7} " +
      " ".join(f"{tot[c]:>16.3e}" for c in counters))
