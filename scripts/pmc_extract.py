#!/usr/bin/env python3
"""Aggregate rocprofv3 PMC counters per kernel (ratio to SQ_BUSY_CYCLES,
normalized by the 16 wave slots per SQ)."""
import collections
import glob
import sys

import sqlite3

db_glob, out_path = sys.argv[1], sys.argv[2]
pdb = sqlite3.connect(glob.glob(db_glob, recursive=True)[0])
cols = [d[1] for d in pdb.execute("PRAGMA table_info(counters_collection)")]
ki = cols.index("kernel_name")
ci = cols.index("counter_name")
vi = cols.index("value")
agg = collections.defaultdict(float)
for r in pdb.execute("SELECT * FROM counters_collection"):
    agg[(r[ki].split("(")[0][:50], r[ci])] += r[vi]
counters = sorted({c for _, c in agg})
with open(out_path, "w") as f:
    for k in sorted({a for a, _ in agg}):
        busy = agg.get((k, "SQ_BUSY_CYCLES"), 0.0)
        parts = [k]
        for c in counters:
            if c == "SQ_BUSY_CYCLES":
                continue
            v = agg.get((k, c), 0.0)
            r = v / busy / 16 if busy else 0.0
            parts.append(f"  {c}: {v:.3e} (ratio {r:.3f})")
        line = "\n".join(parts) + f"\n  SQ_BUSY_CYCLES: {busy:.3e}"
        print(line)
        f.write(line + "\n")
