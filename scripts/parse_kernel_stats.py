import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
for r in rows[:8]:
    short = r["Name"].split("<")[0].split("(")[0][:55]
    ms = float(r["TotalDurationNs"]) / 1e6
    print("%9.1f ms %5s  %s" % (ms, r["Calls"], short))
