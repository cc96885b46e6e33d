import csv, glob, sys
path = sys.argv[1] if len(sys.argv) > 1 else glob.glob("/tmp/prof/**/*kernel_stats*", recursive=True)[0]
rows = list(csv.DictReader(open(path)))
for r in sorted(rows, key=lambda x: -float(x["TotalDurationNs"]))[:10]:
    name = r["Name"][:52]
    print("%-52s %5s avg=%9.1fus tot=%8.2fms" % (name, r["Calls"],
          float(r["AverageNs"])/1e3, float(r["TotalDurationNs"])/1e6))
