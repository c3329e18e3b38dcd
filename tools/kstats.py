#!/usr/bin/env python3
"""Print the top-N kernels of a rocprofv3 kernel_stats.csv."""

import csv
import sys


def main():
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 15
    rows = list(csv.DictReader(open(path)))
    rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
    total = sum(float(r["TotalDurationNs"]) for r in rows)
    print(f"total gpu time: {total / 1e6:.2f} ms over {len(rows)} kernels")
    for r in rows[:top]:
        name = r["Name"]
        if name.startswith("Cijk_") or name.startswith("Custom_Cijk"):
            # rocBLAS kernel: keep the macro-tile token
            toks = [t for t in name.split("_") if t.startswith("MT")]
            name = ("rocblas " + (toks[0] if toks else ""))[:52]
        print(f"{float(r['TotalDurationNs']) / 1e6:9.2f} ms "
              f"{int(r['Calls']):6d} calls {float(r['AverageNs']) / 1e3:9.2f} us  "
              f"{name[:60]}")


if __name__ == "__main__":
    main()
