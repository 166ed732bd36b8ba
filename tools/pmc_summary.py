#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc counter_collection CSV: mean counter value per
kernel.  Usage: pmc_summary.py <csv> [csv...]"""
import collections
import csv
import re
import sys


def main(paths):
    agg = collections.defaultdict(lambda: collections.defaultdict(list))
    for path in paths:
        with open(path) as f:
            for row in csv.DictReader(f):
                name = re.sub(r"[(<].*", "", row.get("Kernel_Name", row.get("Kernel-Name","?"))).strip()
                agg[name][row["Counter_Name"]].append(float(row["Counter_Value"]))
    for kernel, counters in sorted(agg.items()):
        print(kernel[:80])
        for cname, vals in sorted(counters.items()):
            print(f"    {cname:28s} mean={sum(vals) / len(vals):14.2f} "
                  f"n={len(vals)}")


if __name__ == "__main__":
    main(sys.argv[1:])
