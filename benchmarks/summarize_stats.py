import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total kernel time {tot/1e9:.2f} s; {len(rows)} distinct kernels")
for r in rows[:15]:
    pct = float(r["Percentage"]); calls = int(r["Calls"])
    avg = float(r["AverageNs"]) / 1e3
    print(f"{pct:5.1f}% {calls:>8} calls avg {avg:8.1f}us  {r['Name'][:90]}")
