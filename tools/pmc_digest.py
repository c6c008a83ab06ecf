"""Digest rocprofv3 PMC output dirs into a small per-kernel counter table.

Usage: python tools/pmc_digest.py DIR [DIR...]
Scans each dir recursively for *counter_collection.csv and prints, per
(kernel, counter): dispatch count, sum, mean. Kernel names are truncated to
their base (before the '(' of the signature / '.kd' suffix) so the table
stays readable. Streams the csv — safe on large captures.
"""

from __future__ import annotations

import csv
import os
import sys
from collections import defaultdict


def kbase(name: str) -> str:
    name = name.strip('"')
    for sep in ("(", "<"):
        # keep template args: cut only at '(' first
        pass
    cut = name.find("(")
    if cut > 0:
        name = name[:cut]
    if name.endswith(".kd"):
        name = name[:-3]
    return name[-80:]


def digest_dir(d: str):
    acc = defaultdict(lambda: [0, 0.0])  # (kernel, counter) -> [n, sum]
    files = []
    for root, _, fns in os.walk(d):
        for fn in fns:
            if fn.endswith("counter_collection.csv"):
                files.append(os.path.join(root, fn))
    for f in files:
        with open(f, newline="") as fh:
            r = csv.DictReader(fh)
            for row in r:
                k = kbase(row.get("Kernel_Name", "?"))
                c = row.get("Counter_Name", "?")
                try:
                    v = float(row.get("Counter_Value", "0"))
                except ValueError:
                    continue
                e = acc[(k, c)]
                e[0] += 1
                e[1] += v
    return acc, files


def main():
    for d in sys.argv[1:]:
        acc, files = digest_dir(d)
        print(f"== {d} ({len(files)} csv files) ==")
        if not acc:
            print("  (no counter_collection.csv found)")
            continue
        print(f"{'kernel':60s} {'counter':24s} {'n':>6s} {'mean':>14s} {'sum':>16s}")
        for (k, c), (n, s) in sorted(acc.items()):
            print(f"{k:60s} {c:24s} {n:6d} {s / max(n, 1):14.4g} {s:16.6g}")
        print()


if __name__ == "__main__":
    main()
