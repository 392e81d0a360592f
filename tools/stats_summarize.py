#!/usr/bin/env python3
"""Summarize rocprofv3 --stats kernel CSVs into a compact top-N table.

Usage: python tools/stats_summarize.py <dir-or-csv> [topN]
Finds *kernel_stats.csv under the directory, merges them, prints kernels
sorted by total duration with call counts and % of GPU-busy time.
"""

import csv
import glob
import os
import re
import sys


def load(path):
    rows = []
    with open(path, newline="") as f:
        for r in csv.DictReader(f):
            name = r.get("Name") or r.get("NAME") or r.get("KERNEL_NAME")
            if not name:
                continue
            calls = int(float(r.get("Calls") or r.get("CALLS") or 0))
            tot = float(r.get("TotalDurationNs") or r.get("TOTAL_DURATION_NS")
                        or r.get("DurationNs") or 0)
            rows.append((name, calls, tot))
    return rows


def main():
    target = sys.argv[1]
    topn = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    files = ([target] if target.endswith(".csv")
             else glob.glob(os.path.join(target, "**", "*kernel_stats.csv"),
                            recursive=True))
    agg = {}
    for f in files:
        for name, calls, tot in load(f):
            short = re.sub(r"\(.*", "", name.replace(
                "(anonymous namespace)::", "")).strip()[:80]
            c, t = agg.get(short, (0, 0.0))
            agg[short] = (c + calls, t + tot)
    total = sum(t for _, t in agg.values())
    print(f"# {len(files)} stats file(s); GPU-busy total "
          f"{total / 1e6:.3f} ms")
    print(f"{'kernel':<80} {'calls':>7} {'total_ms':>10} {'%':>6}")
    for name, (c, t) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:topn]:
        print(f"{name:<80} {c:>7} {t / 1e6:>10.3f} "
              f"{100 * t / total if total else 0:>6.2f}")


if __name__ == "__main__":
    main()
