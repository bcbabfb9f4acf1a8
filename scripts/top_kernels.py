"""Print the top kernels of a rocprofv3 kernel_stats.csv by total time.

    python scripts/top_kernels.py <kernel_stats.csv> [n]
"""

import csv
import sys

path = sys.argv[1]
n = int(sys.argv[2]) if len(sys.argv) > 2 else 28
rows = list(csv.DictReader(open(path)))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total kernel time: {tot / 1e6:.1f} ms over profiled run")
for r in rows[:n]:
    ms = float(r["TotalDurationNs"]) / 1e6
    print(f"{ms:8.2f} ms {int(r['Calls']):6d}x  {r['Name'][:95]}")
