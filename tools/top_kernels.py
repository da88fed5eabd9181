"""Print top kernels from a rocprofv3 kernel_stats CSV."""
import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"kernel total {tot/1e9:.2f}s")
for r in rows[:12]:
    print(f'{float(r["TotalDurationNs"])/1e9:6.2f}s {int(r["Calls"]):7d} '
          f'{r["Name"][:70]}')
