"""Per-grid-size breakdown of selected kernels from a rocprofv3
kernel-trace CSV (columns Kernel_Name, Start/End or duration, Grid_Size /
Workgroup_Size vary by version -- probe the header).

    python scripts/kernel_breakdown.py <kernel_trace.csv> <substr> [substr...]
"""

import csv
import sys
from collections import defaultdict

path = sys.argv[1]
pats = sys.argv[2:] or ["gn_fwd_reduce", "transpose_kernel", "pw_wgrad"]
rows = list(csv.DictReader(open(path)))
if not rows:
    sys.exit("empty trace")
cols = rows[0].keys()
name_c = next(c for c in cols if "Kernel_Name" in c or c == "Name")
dur_c = next((c for c in cols if "Duration" in c), None)
start_c = next((c for c in cols if "Start" in c), None)
end_c = next((c for c in cols if "End" in c), None)
gx = [c for c in cols if "Grid" in c]
print("columns:", name_c, dur_c or (start_c, end_c), gx)

agg = defaultdict(lambda: [0, 0.0])
for r in rows:
    nm = r[name_c]
    if not any(p in nm for p in pats):
        continue
    if dur_c:
        d = float(r[dur_c])
    else:
        d = float(r[end_c]) - float(r[start_c])
    g = "x".join(r[c] for c in gx)
    key = (nm.split("(")[0][:40], g)
    agg[key][0] += 1
    agg[key][1] += d

for (nm, g), (n, tot) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
    print(f"{tot / 1e6:9.2f} ms {n:6d}x {tot / n / 1e3:8.1f} us/call  grid={g:24s} {nm}")
