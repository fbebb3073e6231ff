import csv, os, sys, glob
path = sys.argv[1]
rows = list(csv.DictReader(open(path)))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
for r in rows[:12]:
    name = r["Name"][:70]
    print("%9.3f ms n=%5s avg=%8.2f us  %s" % (
        float(r["TotalDurationNs"]) / 1e6, r["Calls"],
        float(r["AverageNs"]) / 1e3, name))
