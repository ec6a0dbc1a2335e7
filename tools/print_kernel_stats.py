import csv, sys, glob
fs = []
for a in sys.argv[1:]:
    fs += glob.glob(a, recursive=True)
for f in fs:
    rows = list(csv.DictReader(open(f)))
    rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
    tot = sum(float(r["TotalDurationNs"]) for r in rows)
    print(f, "total %.1f ms" % (tot / 1e6))
    for r in rows[:14]:
        d = float(r["TotalDurationNs"]) / 1e6
        a = float(r["AverageNs"]) / 1e3
        print("  %8.2f ms %6s calls avg %7.2f us  %s"
              % (d, r["Calls"], a, r["Name"][:72]))
