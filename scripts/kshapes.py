# Group a rocprofv3 kernel_trace csv by (kernel, grid) to get per-shape times.
import csv
import sys
from collections import defaultdict

path = sys.argv[1]
agg = defaultdict(lambda: [0, 0.0])
with open(path) as fh:
    for row in csv.DictReader(fh):
        name = row.get("Kernel_Name", row.get("Name", ""))[:60]
        gx = row.get("Workgroup_Count_X") or row.get("Grid_Size_X") or "?"
        gy = row.get("Workgroup_Count_Y") or row.get("Grid_Size_Y") or "?"
        gz = row.get("Workgroup_Count_Z") or row.get("Grid_Size_Z") or "1"
        key = (name, f"{gx}x{gy}x{gz}")
        t0 = row.get("Start_Timestamp")
        t1 = row.get("End_Timestamp")
        if t0 and t1:
            dur = int(t1) - int(t0)
        else:
            dur = int(row.get("Duration", 0) or 0)
        agg[key][0] += 1
        agg[key][1] += dur

rows = sorted(agg.items(), key=lambda kv: -kv[1][1])
for (name, grid), (n, tot) in rows[:40]:
    print(f"{tot / 1e6:9.2f}ms {n:6d}x {tot / n / 1e3:8.1f}us  grid={grid:<14} {name}")
