"""Summarize a rocprofv3 kernel_stats.csv (top-N by total time)."""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: float(r["TotalDurationNs"]), reverse=True)
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total GPU {tot / 1e9:.3f}s over {len(rows)} kernel types")
for r in rows[: int(sys.argv[2]) if len(sys.argv) > 2 else 16]:
    t = float(r["TotalDurationNs"]) / 1e6
    print(f"  {t:7.1f}ms {int(float(r['Calls'])):6d}x {r['Name'][:90]}")
