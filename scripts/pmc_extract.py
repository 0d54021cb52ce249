"""Aggregate rocprofv3 counter_collection CSVs: per (kernel, grid) totals.

Usage: python scripts/pmc_extract.py <dir-with-rocprofv3-output> [counter]
Prints one line per (kernel_name, grid_size, counter): n dispatches, sum,
mean per dispatch — the mean is the per-launch figure pmc_calibration.json
records (FETCH_SIZE needs the gfx950 x2 correction, applied by the reader,
see profiles/pmc_calibration.json note).
"""
import csv
import glob
import os
import sys
from collections import defaultdict


def main():
    root = sys.argv[1]
    files = glob.glob(os.path.join(root, "**", "*counter_collection.csv"),
                      recursive=True)
    if not files:
        print(f"no counter_collection.csv under {root}")
        return
    agg = defaultdict(lambda: [0, 0.0])
    for path in files:
        with open(path) as f:
            for row in csv.DictReader(f):
                name = row.get("Kernel_Name", "?").split("(")[0]
                grid = row.get("Grid_Size", "?")
                ctr = row.get("Counter_Name", "?")
                val = float(row.get("Counter_Value", 0) or 0)
                key = (name, grid, ctr)
                agg[key][0] += 1
                agg[key][1] += val
    for (name, grid, ctr), (n, tot) in sorted(agg.items(),
                                              key=lambda kv: -kv[1][1]):
        print(f"{name:46s} grid={grid:>12s} {ctr:12s} n={n:5d} "
              f"sum={tot:.6e} mean={tot / n:.6e}")


if __name__ == "__main__":
    main()
