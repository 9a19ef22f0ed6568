#!/usr/bin/env python3
"""Report stream-level overlap from a rocprofv3 kernel-trace results.db:
for the data-plane kernels (scale_cast / cast_copy) print their stream and
how much of their run time overlaps dispatches on OTHER streams.

Usage: python benchmarks/overlap_report.py <results.db|dir>
"""
import glob
import os
import sqlite3
import sys


def main(path):
    if os.path.isdir(path):
        path = sorted(glob.glob(os.path.join(path, "**", "*_results.db"),
                                recursive=True))[-1]
    c = sqlite3.connect(path)
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tabs if 'kernel_dispatch' in t][0]
    ks = [t for t in tabs if 'info_kernel_symbol' in t][0]
    rows = list(c.execute(
        f"SELECT s.display_name, d.stream_id, d.start, d.end "
        f"FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id ORDER BY d.start"))
    streams = sorted({r[1] for r in rows})
    print(f"== {path}\nstreams seen: {streams}")
    agg = [r for r in rows if 'scale_cast' in r[0] or 'cast_copy' in r[0]]
    if not agg:
        print("no data-plane kernels in trace")
        return
    agg_streams = sorted({r[1] for r in agg})
    print(f"aggregation kernels: {len(agg)} dispatches on stream(s) {agg_streams}")
    other = [r for r in rows if r[1] not in agg_streams]
    overl = 0
    tot = 0
    for _, _, s0, e0 in agg:
        tot += e0 - s0
        for _, _, s1, e1 in other:
            lo, hi = max(s0, s1), min(e0, e1)
            if hi > lo:
                overl += hi - lo
    print(f"aggregation kernel time: {tot/1e3:.1f} us; "
          f"overlapped with other-stream dispatches: {overl/1e3:.1f} us "
          f"({100*overl/max(tot,1):.0f}%)")
    comp_during = 0
    if agg:
        a0 = min(r[2] for r in agg)
        a1 = max(r[3] for r in agg)
        comp_during = sum(min(e, a1) - max(s, a0)
                          for _, _, s, e in other if e > a0 and s < a1)
    print(f"other-stream kernel time inside the aggregation window: "
          f"{comp_during/1e3:.1f} us")


if __name__ == "__main__":
    main(sys.argv[1])
