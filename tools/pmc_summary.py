#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc CSV into per-kernel mean counter values.

Usage: python tools/pmc_summary.py "<csv-glob>" out.txt
Handles the rocprofv3 counter_collection.csv layout (one row per
dispatch x counter): groups by (kernel, counter), reports mean over
dispatches and dispatch count.
"""

import csv
import glob
import sys
from collections import defaultdict


def main():
    pat, out = sys.argv[1], sys.argv[2]
    files = sorted(glob.glob(pat, recursive=True))
    if not files:
        print(f"no csv matches {pat}", file=sys.stderr)
        sys.exit(1)
    acc = defaultdict(lambda: [0.0, 0])
    kernels = {}
    for f in files:
        with open(f) as fh:
            rd = csv.DictReader(fh)
            cols = {c.lower(): c for c in rd.fieldnames or []}
            kcol = next((cols[c] for c in ("kernel_name", "kernel-name", "name") if c in cols), None)
            ccol = next((cols[c] for c in ("counter_name", "counter-name") if c in cols), None)
            vcol = next((cols[c] for c in ("counter_value", "counter-value", "value") if c in cols), None)
            if not (kcol and ccol and vcol):
                print(f"{f}: unrecognized columns {rd.fieldnames}", file=sys.stderr)
                continue
            for row in rd:
                k = row[kcol].split("(")[0][:80]
                c = row[ccol]
                try:
                    v = float(row[vcol])
                except ValueError:
                    continue
                a = acc[(k, c)]
                a[0] += v
                a[1] += 1
                kernels.setdefault(k, set()).add(c)
    with open(out, "w") as fh:
        for k in sorted(kernels):
            fh.write(f"\n{k}\n")
            for c in sorted(kernels[k]):
                tot, n = acc[(k, c)]
                fh.write(f"  {c:32s} mean={tot / max(n, 1):16.1f}  (n={n})\n")
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
