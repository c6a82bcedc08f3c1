#!/usr/bin/env python3
"""Aggregate a rocprofv3 counter_collection CSV (one row per dispatch x
counter) into one row per kernel: sum per counter + dispatch count.
Usage: pmc_summarize.py <counter_collection.csv> [> summary.csv]"""
import csv
import sys


def main(path):
    agg = {}   # kernel -> {counter: sum}
    counts = {}
    with open(path) as f:
        for row in csv.DictReader(f):
            k = row.get("Kernel_Name") or row.get("Kernel Name") or ""
            c = row.get("Counter_Name") or row.get("Counter Name") or ""
            v = float(row.get("Counter_Value") or row.get("Counter Value") or 0)
            d = agg.setdefault(k, {})
            d[c] = d.get(c, 0.0) + v
            key = (k, row.get("Dispatch_Id") or row.get("Dispatch_ID"))
            counts.setdefault(k, set()).add(key[1])
    counters = sorted({c for d in agg.values() for c in d})
    w = csv.writer(sys.stdout)
    w.writerow(["Kernel", "Dispatches"] + counters)
    for k in sorted(agg, key=lambda k: -agg[k].get("SQ_WAVE_CYCLES", 0)):
        w.writerow([k[:130], len(counts[k])] +
                   [f"{agg[k].get(c, 0):.0f}" for c in counters])


if __name__ == "__main__":
    main(sys.argv[1])
