#!/usr/bin/env python3
"""Summarize a rocprofv3 *_kernel_stats.csv: top kernels by total time."""

import csv
import glob
import sys


def main():
    pat = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/prof/*kernel_stats.csv"
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 25
    files = glob.glob(pat)
    if not files:
        print("no kernel_stats.csv found for", pat)
        return
    rows = list(csv.DictReader(open(files[0])))
    rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
    tot = sum(float(r["TotalDurationNs"]) for r in rows)
    print("total GPU s:", round(tot / 1e9, 3), " kernels:", len(rows))
    for r in rows[:n]:
        ms = float(r["TotalDurationNs"]) / 1e6
        pct = float(r["Percentage"])
        print(f"{ms:9.1f}ms {pct:5.2f}% {int(r['Calls']):6d}x  "
              f"{r['Name'][:110]}")


if __name__ == "__main__":
    main()
