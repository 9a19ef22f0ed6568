#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a per-kernel time table.

Usage: python benchmarks/profstats.py <results.db|dir> [top_n]
"""
import glob
import os
import re
import sqlite3
import sys


def summarize(db, top=30):
    c = sqlite3.connect(db)
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tabs if 'kernel_dispatch' in t][0]
    ks = [t for t in tabs if 'info_kernel_symbol' in t][0]
    rows = list(c.execute(
        f"SELECT s.display_name, COUNT(*), SUM(d.end - d.start) "
        f"FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id "
        f"GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"))
    total = sum(r[2] for r in rows) or 1
    out = [f"total kernel time: {total/1e6:.3f} ms over "
           f"{sum(r[1] for r in rows)} dispatches"]
    out.append(f"{'%':>6} {'ms':>10} {'calls':>7}  name")
    for name, n, t in rows[:top]:
        short = re.sub(r'\(.*', '', name)[:100]
        out.append(f"{100*t/total:6.2f} {t/1e6:10.3f} {n:7d}  {short}")
    return "\n".join(out)


if __name__ == "__main__":
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    if os.path.isdir(path):
        dbs = sorted(glob.glob(os.path.join(path, "**", "*_results.db"),
                               recursive=True))
        path = dbs[-1]
    print(f"== {path}")
    print(summarize(path, top))
