"""Aggregate a rocprofv3 counter_collection.csv into per-kernel sums.

Usage: python scripts/aggregate_pmc.py <csv> [<out_csv>]

Groups by (kernel, counter) and prints total counter values (units as
rocprofv3 reports them; FETCH_SIZE is kilobytes fetched past TCC).
SQ_INSTS_MFMA / SQ_BUSY_CYCLES gives an MFMA-issue-density proxy per
kernel.
"""

import csv
import sys
from collections import defaultdict


def main():
    path = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else None
    sums = defaultdict(float)
    dispatches = defaultdict(int)
    with open(path, newline="") as f:
        reader = csv.DictReader(f)
        kcol = ccol = vcol = None
        for row in reader:
            if kcol is None:
                keys = {k.lower(): k for k in row}
                kcol = keys.get("kernel_name")
                ccol = keys.get("counter_name")
                vcol = keys.get("counter_value")
            name = row[kcol].replace("(anonymous namespace)::", "")
            name = name.split("(")[0][:80]
            sums[(name, row[ccol])] += float(row[vcol])
            if row[ccol].endswith("WAVES"):
                dispatches[name] += 1
    lines = [("kernel", "counter", "total")]
    for (kern, ctr), val in sorted(sums.items(),
                                   key=lambda kv: -kv[1]):
        lines.append((kern, ctr, f"{val:.6g}"))
    text = "\n".join(",".join(map(str, ln)) for ln in lines)
    print(text)
    if out:
        with open(out, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
