# Summarize a rocprofv3 kernel_stats csv (top-N by total time).
import csv
import sys

path = sys.argv[1]
top = int(sys.argv[2]) if len(sys.argv) > 2 else 25
rows = list(csv.DictReader(open(path)))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total {tot / 1e6:.1f}ms over {sum(int(r['Calls']) for r in rows)} dispatches")
for r in rows[:top]:
    t = float(r["TotalDurationNs"]) / 1e6
    n = int(r["Calls"])
    avg = float(r["AverageNs"]) / 1e3
    print(f"{t:9.2f}ms {n:7d}x {avg:9.1f}us  {r['Name'][:80]}")
