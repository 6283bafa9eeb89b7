"""Print per-kernel averages from a rocprofv3 kernel_stats CSV."""
import csv
import sys

for row in csv.DictReader(open(sys.argv[1])):
    n = row["Name"]
    if len(sys.argv) < 3 or any(k in n for k in sys.argv[2].split(",")):
        calls = int(row["Calls"])
        avg = float(row["AverageNs"]) / 1e3
        tot = float(row["TotalDurationNs"]) / 1e6
        print(f"{n[:56]:58s} {calls:4d} calls  {avg:9.1f} us avg  {tot:8.2f} ms tot")
