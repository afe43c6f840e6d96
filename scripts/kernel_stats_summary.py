"""Summarize a rocprofv3 kernel_stats.csv (top-N by total time)."""
import csv, sys

path = sys.argv[1]
rows = list(csv.DictReader(open(path)))
key = "TotalDurationNs" if rows and "TotalDurationNs" in rows[0] else "TOTAL_DURATION_NS"
rows.sort(key=lambda r: -float(r[key]))
tot = sum(float(r[key]) for r in rows)
print(f"total kernel time {tot/1e9:.2f}s; top 15:")
for r in rows[:15]:
    name = (r.get("Name") or r.get("NAME", "?"))[:80]
    calls = int(r.get("Calls") or r.get("CALLS", 0))
    print(f"{float(r[key])/tot*100:5.1f}%  {calls:6d}x  {name}")
