#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc counter_collection.csv: per-kernel stall/
issue/VALU-per-MFMA ratios.  Usage: pmc_summary.py <glob-dir>"""
import csv, glob, sys, collections

rows = collections.defaultdict(lambda: collections.defaultdict(float))
for f in glob.glob(sys.argv[1] + "/*counter_collection.csv"):
    for r in csv.DictReader(open(f)):
        name = r.get("Kernel_Name", "")[:44]
        rows[name][r.get("Counter_Name")] += float(r.get("Counter_Value", 0))
for name, d in sorted(rows.items(), key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", 0)):
    wc = d.get("SQ_WAVE_CYCLES", 0)
    if wc < 1e7:
        continue
    mfma = max(d.get("SQ_INSTS_MFMA", 0), 1.0)
    print(f"{name:44s} wc={wc/1e9:6.2f}G stall%={100*d.get('SQ_WAIT_ANY',0)/wc:5.1f} "
          f"instl%={100*d.get('SQ_WAIT_INST_ANY',0)/wc:5.1f} "
          f"act%={100*d.get('SQ_ACTIVE_INST_ANY',0)/wc:5.1f} "
          f"valu/mfma={d.get('SQ_INSTS_VALU',0)/mfma:6.2f} "
          f"lds/mfma={d.get('SQ_INSTS_LDS',0)/mfma:5.2f} "
          f"conf%={100*d.get('SQ_LDS_BANK_CONFLICT',0)/wc:4.1f}")
