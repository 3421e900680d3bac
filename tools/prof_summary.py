"""Summarize rocprofv3 kernel_stats CSV: top kernels by total time."""
import csv
import glob
import sys

pattern = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/*kernel_stats*.csv"
files = glob.glob(pattern)
if not files:
    sys.exit(f"no kernel stats files match {pattern}")
f = sorted(files)[-1]
rows = list(csv.DictReader(open(f)))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
print(f"# {f}")
print(f"total kernel time {tot/1e9:.3f}s over {sum(int(r['Calls']) for r in rows)} launches")
for r in rows[:20]:
    pct = float(r["TotalDurationNs"]) / tot * 100
    print(
        "%5.1f%% %6dx %9.1fus  %s"
        % (pct, int(r["Calls"]), float(r["AverageNs"]) / 1e3, r["Name"][:100])
    )
