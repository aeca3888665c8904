#!/usr/bin/env python3
"""Summarize rocprofv3 PMC counter CSVs per kernel (wait/active split)."""
import collections
import csv
import glob
import sys

pat = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/*counter*.csv"
files = glob.glob(pat, recursive=True)
rows = []
for f in files:
    rows += list(csv.DictReader(open(f)))
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for r in rows:
    agg[r["Kernel_Name"][:30]][r["Counter_Name"]] += float(r["Counter_Value"])
for n, c in sorted(agg.items(), key=lambda kv: -kv[1]["SQ_WAVE_CYCLES"]):
    wc = c["SQ_WAVE_CYCLES"]
    if not wc:
        continue
    act = 100 * c["SQ_ACTIVE_INST_ANY"] / wc
    wait = 100 * c["SQ_WAIT_ANY"] / wc
    wi = 100 * c["SQ_WAIT_INST_ANY"] / wc
    conf = c["SQ_LDS_BANK_CONFLICT"] / 1e6
    print(f"{n:30s} ACTIVE {act:5.1f}%  WAIT {wait:5.1f}%  "
          f"WAITINST {wi:5.1f}%  BANKCONF {conf:7.0f}M")
