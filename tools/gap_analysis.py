#!/usr/bin/env python3
"""GPU busy/gap analysis from a rocprofv3 kernel-trace CSV.

Merges kernel intervals and reports wall span vs busy time (launch-gap
diagnosis). Run on the GPU box against the *_kernel_trace.csv before
deleting it.
"""

import csv
import glob
import sys


def main():
    pat = sys.argv[1] if len(sys.argv) > 1 else \
        "gpurun_out/prof/*kernel_trace.csv"
    files = glob.glob(pat)
    if not files:
        print("no kernel_trace.csv for", pat)
        return
    intervals = []
    with open(files[0]) as f:
        r = csv.DictReader(f)
        start_key = end_key = None
        for row in r:
            if start_key is None:
                keys = {k.lower(): k for k in row}
                start_key = keys.get("start_timestamp")
                end_key = keys.get("end_timestamp")
                if not start_key:
                    print("columns:", list(row))
                    return
            intervals.append((int(row[start_key]), int(row[end_key])))
    intervals.sort()
    busy = 0
    cur_s, cur_e = intervals[0]
    n_gaps = 0
    gap_time = 0
    for s, e in intervals[1:]:
        if s > cur_e:
            busy += cur_e - cur_s
            n_gaps += 1
            gap_time += s - cur_e
            cur_s, cur_e = s, e
        else:
            cur_e = max(cur_e, e)
    busy += cur_e - cur_s
    span = intervals[-1][1] - intervals[0][0]
    print(f"kernels: {len(intervals)}")
    print(f"span:  {span / 1e6:.1f} ms")
    print(f"busy:  {busy / 1e6:.1f} ms ({100 * busy / span:.1f}%)")
    print(f"gaps:  {n_gaps} totalling {gap_time / 1e6:.1f} ms "
          f"(avg {gap_time / max(n_gaps, 1) / 1e3:.1f} us)")
    # Largest 15 gaps.
    gaps = []
    cur_e = intervals[0][1]
    for s, e in intervals[1:]:
        if s > cur_e:
            gaps.append((s - cur_e, cur_e))
        cur_e = max(cur_e, e)
    gaps.sort(reverse=True)
    print("largest gaps (us, at ms-into-trace):")
    t0 = intervals[0][0]
    for g, at in gaps[:15]:
        print(f"  {g / 1e3:9.1f} us at {(at - t0) / 1e6:9.1f} ms")


if __name__ == "__main__":
    main()
