"""Print the top kernels of a rocprofv3 kernel_stats CSV."""

import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
total = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total kernel time {total/1e6:.1f} ms over run")
for r in rows[: int(sys.argv[2]) if len(sys.argv) > 2 else 12]:
    name = r["Name"][:100]
    print(f"{float(r['TotalDurationNs'])/1e6:9.2f} ms {int(r['Calls']):5d}x  {name}")
