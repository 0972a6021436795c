"""Summarize a rocprofv3 kernel_stats.csv: top-N kernels by total time.

  python -m bigslice_amd.tools.ksum <kernel_stats.csv> [N]
"""

import csv
import sys


def main():
    path = sys.argv[1]
    topn = int(sys.argv[2]) if len(sys.argv) > 2 else 15
    rows = list(csv.DictReader(open(path)))
    rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
    for r in rows[:topn]:
        name = r["Name"][:72]
        tot = float(r["TotalDurationNs"]) / 1e6
        avg = float(r["AverageNs"]) / 1e3
        print(f"{name:72s} calls={r['Calls']:>5s} tot_ms={tot:9.2f} "
              f"avg_us={avg:9.1f} pct={r['Percentage'][:5]}")


if __name__ == "__main__":
    main()
