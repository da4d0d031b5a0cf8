#!/usr/bin/env python3
"""Post-process rocprofv3 --pmc SQ wave-cycle counters into the
per-kernel %WAIT_ANY / %WAIT_INST / %ACTIVE decomposition
(profiles/*_pmc_wavecycle_decomposition format). Ratios only — PMC
serialization inflates absolute durations.
usage: pmc_wave_post.py <counter_collection.csv>"""
import csv
import sys
from collections import defaultdict

acc = defaultdict(lambda: defaultdict(float))
with open(sys.argv[1]) as f:
    for row in csv.DictReader(f):
        kn = (row.get("Kernel_Name") or "").split("(")[0] \
            .replace("void ", "").strip()
        cn = row.get("Counter_Name") or ""
        cv = float(row.get("Counter_Value") or 0)
        if kn and cn:
            acc[kn][cn] += cv

print(f"{'kernel':40s} {'%WAIT_ANY':>9s} {'%WAIT_INST':>10s} "
      f"{'%ACTIVE':>8s}")
for k in sorted(acc, key=lambda k: -acc[k].get("SQ_WAVE_CYCLES", 0)):
    wc = acc[k].get("SQ_WAVE_CYCLES", 0)
    if wc < 1e6:
        continue
    wa = acc[k].get("SQ_WAIT_ANY", 0) / wc * 100
    wi = acc[k].get("SQ_WAIT_INST_ANY", 0) / wc * 100
    ac = acc[k].get("SQ_ACTIVE_INST_ANY", 0) / wc * 100
    print(f"{k[:40]:40s} {wa:9.1f} {wi:10.1f} {ac:8.1f}")
