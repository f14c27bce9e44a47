#!/usr/bin/env python3
"""Summarize the steady-state tail of a rocprofv3 kernel trace.

Usage: analyze_ktrace.py TRACE.csv [window_ms] [steps]

Takes the last `window_ms` (default 150) of the trace — i.e. the timed
bench region, after MIOpen find and warmup — groups kernels by name, and
prints a per-step cost table (assuming `steps` steps in the window).
"""

import csv
import sys
from collections import defaultdict


def main():
    path = sys.argv[1]
    window_ms = float(sys.argv[2]) if len(sys.argv) > 2 else 150.0
    steps = int(sys.argv[3]) if len(sys.argv) > 3 else None

    rows = []
    with open(path) as f:
        for r in csv.DictReader(f):
            start = float(r["Start_Timestamp"])
            end = float(r["End_Timestamp"])
            rows.append((start, end, r["Kernel_Name"]))
    rows.sort()
    t_end = max(e for _, e, _ in rows)
    t_cut = t_end - window_ms * 1e6

    agg = defaultdict(lambda: [0, 0.0])  # name -> [calls, ns]
    busy_ns = 0.0
    span_start = None
    prev_end = None
    gap_ns = 0.0
    for s, e, name in rows:
        if e < t_cut:
            continue
        if span_start is None:
            span_start = s
        agg[name][0] += 1
        agg[name][1] += e - s
        if prev_end is not None and s > prev_end:
            gap_ns += s - prev_end
        prev_end = max(prev_end or e, e)
        busy_ns += e - s

    span_ns = t_end - span_start
    total_ns = sum(v[1] for v in agg.values())
    print(f"window: {span_ns/1e6:.1f} ms, kernel-busy {busy_ns/1e6:.1f} ms "
          f"({100*busy_ns/span_ns:.0f}%), gaps {gap_ns/1e6:.1f} ms, "
          f"{sum(v[0] for v in agg.values())} launches")
    if steps:
        print(f"per-step: {span_ns/1e6/steps:.3f} ms wall, "
              f"{busy_ns/1e6/steps:.3f} ms busy, "
              f"{sum(v[0] for v in agg.values())/steps:.0f} launches")
    print(f"{'calls':>7} {'tot_ms':>8} {'avg_us':>8} {'%':>5}  name")
    for name, (calls, ns) in sorted(
        agg.items(), key=lambda kv: -kv[1][1]
    )[:30]:
        print(f"{calls:>7} {ns/1e6:>8.2f} {ns/1e3/calls:>8.1f} "
              f"{100*ns/total_ns:>5.1f}  {name[:95]}")


if __name__ == "__main__":
    main()
