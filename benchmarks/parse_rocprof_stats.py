#!/usr/bin/env python3
"""Render a rocprofv3 kernel-stats CSV as the profiles/ markdown table.

Usage: python benchmarks/parse_rocprof_stats.py <kernel_stats.csv> [top_n]
"""

import csv
import sys


def main() -> None:
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 25
    rows = []
    with open(path, newline="") as f:
        for r in csv.DictReader(f):
            # rocprofv3 csv stats columns: "Name", "Calls",
            # "TotalDurationNs", "AverageNs", "Percentage", ...
            name = r.get("Name") or r.get("NAME") or ""
            calls = int(r.get("Calls") or r.get("CALLS") or 0)
            tot_ns = float(r.get("TotalDurationNs") or r.get("DurationNs") or 0)
            rows.append((tot_ns, calls, name))
    rows.sort(reverse=True)
    total_ms = sum(t for t, _, _ in rows) / 1e6
    print("| total_ms | calls | avg_us | kernel |")
    print("|---|---|---|---|")
    for tot_ns, calls, name in rows[:top]:
        avg_us = tot_ns / max(1, calls) / 1e3
        print(f"| {tot_ns / 1e6:.2f} | {calls} | {avg_us:.1f} | `{name[:80]}` |")
    print(f"\nTOTAL kernel time: {total_ms:.1f} ms")


if __name__ == "__main__":
    main()
