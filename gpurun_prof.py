import csv, glob
rows = list(csv.DictReader(open(glob.glob("/tmp/prof/**/*kernel_stats*", recursive=True)[0])))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
for r in sorted(rows, key=lambda r: -float(r["TotalDurationNs"]))[:7]:
    n = r["Name"].split("(")[0][:58]
    print("%-60s %3d avg %8.1fus %5.1f%%" % (n, int(r["Calls"]), float(r["AverageNs"])/1000, 100*float(r["TotalDurationNs"])/tot))
