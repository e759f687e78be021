#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc csv output: per-kernel mean counter values and
mean dispatch time.  Usage: python pmc_summarize.py <counter_csv> [...]"""
import csv
import sys
from collections import defaultdict


def main():
    # (kernel, counter) -> [sum, n]; kernel -> [time_sum_ns, n_dispatch]
    vals = defaultdict(lambda: [0.0, 0])
    times = defaultdict(lambda: [0.0, set()])
    for path in sys.argv[1:]:
        with open(path) as f:
            for row in csv.DictReader(f):
                name = row["Kernel_Name"].split("(")[0]
                if "rocclr" in name:
                    continue
                v = vals[(name, row["Counter_Name"])]
                v[0] += float(row["Counter_Value"])
                v[1] += 1
                t = times[name]
                key = (path, row["Dispatch_Id"])
                if key not in t[1]:
                    t[1].add(key)
                    t[0] += (int(row["End_Timestamp"])
                             - int(row["Start_Timestamp"]))
    for name in sorted(times):
        nt = len(times[name][1])
        us = times[name][0] / nt / 1e3
        print(f"\n{name}  ({nt} dispatches, avg {us:.1f} us)")
        counters = sorted(c for (k, c) in vals if k == name)
        for c in counters:
            s, n = vals[(name, c)]
            per = s / max(1, nt)  # counter rows are per-dispatch sums already
            if "SIZE" in c:
                # TCC_*_SIZE counts 32B units
                print(f"  {c:<28} {per * 32 / 1e6:12.1f} MB/dispatch")
            else:
                print(f"  {c:<28} {per:12.3e} /dispatch")
        # derived: MFMA busy fraction if both present
        w = vals.get((name, "SQ_WAVE_CYCLES"))
        m = vals.get((name, "SQ_VALU_MFMA_BUSY_CYCLES"))
        a = vals.get((name, "SQ_WAIT_ANY"))
        ia = vals.get((name, "SQ_WAIT_INST_ANY"))
        if w and w[0] > 0:
            # SQ_WAVE_CYCLES etc. count quad-cycles; MFMA_BUSY counts cycles
            if m:
                print(f"  {'-> MFMA busy / wave cyc':<28} "
                      f"{m[0] / (4 * w[0]) * 100:11.1f} %")
            if a:
                print(f"  {'-> WAIT_ANY / wave cyc':<28} "
                      f"{a[0] / w[0] * 100:11.1f} %")
            if ia:
                print(f"  {'-> WAIT_INST / wave cyc':<28} "
                      f"{ia[0] / w[0] * 100:11.1f} %")


if __name__ == "__main__":
    main()
