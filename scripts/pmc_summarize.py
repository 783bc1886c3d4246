"""Summarize a rocprofv3 counter_collection.csv into per-kernel
MFMA/wave, wait/wave and LDS conflicts/dispatch (the round-2 PMC
report format in profiles/)."""
import csv
import sys
from collections import defaultdict

path = sys.argv[1]
agg = defaultdict(lambda: defaultdict(float))
disp = defaultdict(set)
with open(path) as f:
    for row in csv.DictReader(f):
        k = row["Kernel_Name"][:60]
        agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
        disp[k].add(row["Dispatch_Id"])
for k in sorted(agg, key=lambda k: -agg[k].get("SQ_VALU_MFMA_BUSY_CYCLES", 0)):
    c = agg[k]
    waves = c.get("SQ_WAVE_CYCLES", 0) or 1
    mfma = c.get("SQ_VALU_MFMA_BUSY_CYCLES", 0) / waves * 100
    wait = c.get("SQ_WAIT_ANY", 0) / waves * 100
    conf = c.get("SQ_LDS_BANK_CONFLICT", 0) / max(1, len(disp[k]))
    print(f"{k:<46} MFMA/wave {mfma:5.1f}%  wait/wave {wait:5.1f}%  "
          f"conflicts/disp {conf:.2e}")
