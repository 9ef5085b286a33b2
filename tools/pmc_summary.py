#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc counter_collection.csv into a compact
per-kernel table (the raw CSV has one row per dispatch x counter and
easily exceeds the 64 MiB gpurun pull budget — summarize ON the box,
commit the summary).

Usage: python tools/pmc_summary.py 'gpurun_out/r2_pmc/*counter_collection.csv' [top_n]

Emits per kernel: dispatches, per-counter totals, and the derived
MfmaUtil% = SQ_VALU_MFMA_BUSY_CYCLES / (SQ_WAVE_CYCLES/4) when both
counters are present (WAVE_CYCLES counts quad-cycles —
MI355X_MICROARCH.md s_memtime row).
"""
import csv
import glob
import sys
from collections import defaultdict


def main():
    pat = sys.argv[1] if len(sys.argv) > 1 else \
        "gpurun_out/r2_pmc/*counter_collection.csv"
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    files = sorted(glob.glob(pat))
    if not files:
        print(f"no files match {pat}")
        return 1
    # kernel -> counter -> sum; kernel -> dispatch ids
    table = defaultdict(lambda: defaultdict(float))
    dispatches = defaultdict(set)
    for f in files:
        with open(f, newline="") as fh:
            rd = csv.DictReader(fh)
            for row in rd:
                name = (row.get("Kernel_Name") or row.get("Name")
                        or "?")[:100]
                ctr = row.get("Counter_Name") or row.get("Counter") or "?"
                val = float(row.get("Counter_Value") or
                            row.get("Value") or 0)
                table[name][ctr] += val
                did = row.get("Dispatch_Id") or row.get("Dispatch_ID")
                if did is not None:
                    dispatches[name].add((f, did))
    counters = sorted({c for v in table.values() for c in v})
    order = sorted(table,
                   key=lambda k: -table[k].get("SQ_WAVE_CYCLES", 0))
    print("counters:", ", ".join(counters))
    for name in order[:top]:
        v = table[name]
        n = len(dispatches[name]) or 1
        line = f"{name[:80]:80s} n={n:5d}"
        for c in counters:
            line += f" {c.replace('SQ_', '')}={v.get(c, 0):.3e}"
        mfma = v.get("SQ_VALU_MFMA_BUSY_CYCLES")
        wave = v.get("SQ_WAVE_CYCLES")
        if mfma and wave:
            line += f" MfmaUtil={100.0 * mfma / (wave / 4.0):5.1f}%"
        conf = v.get("SQ_LDS_BANK_CONFLICT")
        idx = v.get("SQ_LDS_IDX_ACTIVE")
        if conf is not None and idx:
            line += f" LdsConflict={100.0 * conf / idx:5.1f}%"
        print(line)
    return 0


if __name__ == "__main__":
    sys.exit(main())
