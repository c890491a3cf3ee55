"""Aggregate a rocprofv3 --pmc counter_collection.csv by kernel.

Usage: python tools/pmc_stats.py gpurun_out/pmc/*_counter_collection.csv
Prints a markdown table: kernel x counter -> sum over dispatches (and the
dispatch count). FETCH_SIZE on gfx950 reports ~1/2 of the true bytes of
wide coalesced streaming reads (MI355X_MICROARCH.md §HBM) — compare
ratios, or double before quoting absolutes for 16B/lane access.
"""

import csv
import sys
from collections import defaultdict


def main(paths):
    # (kernel, counter) -> [sum, dispatches]
    agg = defaultdict(lambda: [0.0, 0])
    for path in paths:
        with open(path) as f:
            rows = list(csv.DictReader(f))
        for r in rows:
            kern = (r.get("Kernel_Name") or r.get("kernel_name") or "?")
            kern = kern.split("(")[0].strip('"')[:60]
            cname = r.get("Counter_Name") or r.get("counter_name") or "?"
            val = float(r.get("Counter_Value") or r.get("counter_value") or 0)
            a = agg[(kern, cname)]
            a[0] += val
            a[1] += 1
    counters = sorted({c for (_, c) in agg})
    kernels = sorted({k for (k, _) in agg},
                     key=lambda k: -max(agg.get((k, c), [0, 0])[0]
                                        for c in counters))
    print("| kernel | dispatches | " + " | ".join(counters) + " |")
    print("|---" * (len(counters) + 2) + "|")
    for k in kernels:
        n = max(agg.get((k, c), [0, 0])[1] for c in counters)
        cells = []
        for c in counters:
            v = agg.get((k, c), [0, 0])[0]
            cells.append(f"{v:.4g}")
        print(f"| {k} | {n} | " + " | ".join(cells) + " |")


if __name__ == "__main__":
    main(sys.argv[1:])
