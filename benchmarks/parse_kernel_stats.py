"""Format a rocprofv3 kernel-stats CSV into a one-line-per-kernel breakdown.

Usage: python benchmarks/parse_kernel_stats.py <dir-or-csv> [topN]
Finds the *kernel_stats*.csv under a directory (rocprofv3 -d output) and
prints total-time-sorted rows.  Column names vary slightly across rocprofv3
versions, so both "TotalDurationNs" and "DurationNs"-style headers work.
"""

import csv
import glob
import os
import sys


def find_csv(path):
    if os.path.isfile(path):
        return path
    for pat in ("*kernel_stats*.csv", "*_stats*.csv", "*.csv"):
        hits = sorted(glob.glob(os.path.join(path, "**", pat), recursive=True))
        hits = [h for h in hits if "domain" not in os.path.basename(h)]
        if hits:
            return hits[0]
    raise SystemExit(f"no stats csv under {path}")


def pick(row, *names):
    for n in names:
        for k in row:
            if k.strip().lower().replace("_", "").replace(" ", "") == n:
                return row[k]
    raise KeyError(f"{names} not in {list(row)}")


def main():
    path = find_csv(sys.argv[1])
    topn = int(sys.argv[2]) if len(sys.argv) > 2 else 14
    rows = list(csv.DictReader(open(path)))
    parsed = []
    for r in rows:
        name = pick(r, "name", "kernelname")
        dur = float(pick(r, "totaldurationns", "durationns", "totalduration"))
        n = int(pick(r, "calls", "count"))
        parsed.append((dur, n, name))
    parsed.sort(reverse=True)
    tot = sum(d for d, _, _ in parsed) or 1.0
    for d, n, name in parsed[:topn]:
        print(f"{d / 1e6:9.1f}ms {d / tot * 100:5.1f}% n={n:5d} "
              f"avg={d / n / 1e3:8.1f}us  {name[:80]}")


if __name__ == "__main__":
    main()
