#!/usr/bin/env python3
"""Print the top-N kernels of a rocprofv3 *_kernel_stats.csv by total time."""
import csv
import sys

path = sys.argv[1]
n = int(sys.argv[2]) if len(sys.argv) > 2 else 18
rows = list(csv.DictReader(open(path)))


def col(r, *names):
    for k in names:
        if k in r:
            return r[k]
    raise KeyError(names)


rows.sort(key=lambda r: -float(col(r, "TotalDurationNs", "DurationNs")))
tot = sum(float(col(r, "TotalDurationNs", "DurationNs")) for r in rows)
print(f"total GPU kernel time: {tot / 1e6:.1f} ms")
for r in rows[:n]:
    t = float(col(r, "TotalDurationNs", "DurationNs"))
    calls = col(r, "Calls", "CallCount")
    avg = float(col(r, "AverageNs", "AvgNs", "AverageDurationNs"))
    print(f"{t / 1e6:9.2f} ms {t / tot * 100:5.1f}% calls={calls:>7} "
          f"avg={avg / 1e3:9.1f} us  {col(r, 'Name')[:80]}")
