"""Summarize a rocprofv3 kernel_stats.csv: ms/step over N steps."""
import csv, sys
f, steps = sys.argv[1], float(sys.argv[2])
rows = list(csv.DictReader(open(f)))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"total gpu {tot/1e6:.1f} ms -> {tot/steps/1e6:.3f} ms/step over {steps:g} steps")
for r in rows[:30]:
    ms = float(r["TotalDurationNs"]) / steps / 1e6
    print(f'{ms:8.4f} ms/step n={int(r["Calls"])/steps:7.1f} avg={float(r["AverageNs"])/1e3:8.2f}us  {r["Name"][:84]}')
