import csv, sys
rows = [r for r in csv.DictReader(open(sys.argv[1]))]
rows.sort(key=lambda r: -float(r['TotalDurationNs']))
for r in rows[:12]:
    t = float(r['TotalDurationNs']) / 1e6
    a = float(r['AverageNs']) / 1e3
    print("%8.2fms %4sc %7.1fus %s" % (t, r['Calls'], a, r['Name'][:72]))
