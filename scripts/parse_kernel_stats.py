#!/usr/bin/env python3
"""Summarize a rocprofv3 --stats kernel CSV: total ms, calls, name.

Usage: python scripts/parse_kernel_stats.py <kernel_stats.csv> [top_n]
"""
import csv
import sys


def main():
    if len(sys.argv) < 2:
        print(__doc__)
        return 1
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 10
    rows = list(csv.DictReader(open(sys.argv[1])))
    for r in rows[:top]:
        short = r["Name"].split("<")[0].split("(")[0][:60]
        ms = float(r["TotalDurationNs"]) / 1e6
        pct = float(r.get("Percentage", 0))
        print("%9.1f ms %5.1f%% %6s calls  %s"
              % (ms, pct, r["Calls"], short))
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
