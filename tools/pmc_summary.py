"""Aggregate a rocprofv3 --pmc counter CSV: mean counter value per kernel.

Usage: python tools/pmc_summary.py <counter_collection.csv-or-dir>
"""

import csv
import glob
import os
import sys
from collections import defaultdict


def main(path):
    if os.path.isdir(path):
        files = glob.glob(os.path.join(path, "**", "*counter*.csv"),
                          recursive=True)
        if not files:
            files = glob.glob(os.path.join(path, "**", "*.csv"),
                              recursive=True)
        path = sorted(files)[-1]
    print(f"# {path}")
    agg = defaultdict(lambda: defaultdict(float))
    cnt = defaultdict(lambda: defaultdict(int))
    with open(path) as f:
        r = csv.DictReader(f)
        for row in r:
            k = row.get("Kernel_Name") or row.get("Kernel-Name") or ""
            c = row.get("Counter_Name") or row.get("Counter-Name") or ""
            v = float(row.get("Counter_Value")
                      or row.get("Counter-Value") or 0)
            k = k.split("(")[0][:40]
            agg[k][c] += v
            cnt[k][c] += 1
    for k in sorted(agg, key=lambda k: -sum(agg[k].values())):
        rows = [
            f"{c}: total {agg[k][c]:.3e} avg {agg[k][c] / max(cnt[k][c], 1):.3e} (n={cnt[k][c]})"
            for c in sorted(agg[k])
        ]
        print(f"== {k}")
        for rr in rows:
            print("   ", rr)


if __name__ == "__main__":
    main(sys.argv[1])
