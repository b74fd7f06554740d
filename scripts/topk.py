import csv
import sys

rows = list(csv.DictReader(open(sys.argv[1])))
key = "TotalDurationNs" if rows and "TotalDurationNs" in rows[0] else "DurationNs"
rows.sort(key=lambda r: -float(r[key]))
total = sum(float(r[key]) for r in rows)
for r in rows[: int(sys.argv[2]) if len(sys.argv) > 2 else 16]:
    ms = float(r[key]) / 1e6
    pct = float(r[key]) / total * 100
    print("%9.1f ms %5.1f%% %6s  %s" % (ms, pct, r.get("Calls", "?"), r["Name"][:90]))
