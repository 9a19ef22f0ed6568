#!/usr/bin/env python3
"""Per-kernel PMC summary from a rocprofv3 --pmc results.db.

Usage: python benchmarks/pmcstats.py <db|dir> [kernel-substring ...]
"""
import glob
import os
import sqlite3
import sys


def main(path, kernels):
    if os.path.isdir(path):
        path = sorted(glob.glob(os.path.join(path, "**", "*_results.db"),
                                recursive=True))[-1]
    c = sqlite3.connect(path)
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    pe = [t for t in tabs if 'pmc_event' in t][0]
    pi = [t for t in tabs if 'info_pmc' in t][0]
    kd = [t for t in tabs if 'kernel_dispatch' in t][0]
    ks = [t for t in tabs if 'info_kernel_symbol' in t][0]
    # counter id -> name
    names = {r[0]: r[1] for r in c.execute(f"SELECT id, name FROM {pi}")}
    # per-dispatch counters joined to kernel names (event_id == dispatch's
    # event_id); sum PMC instances per dispatch
    q = f"""
      SELECT s.display_name, p.pmc_id, SUM(p.value), COUNT(DISTINCT d.id),
             SUM(d.end - d.start)
      FROM {pe} p
      JOIN {kd} d ON p.event_id = d.event_id
      JOIN {ks} s ON d.kernel_id = s.id
      GROUP BY s.display_name, p.pmc_id
    """
    agg = {}
    for name, pmc, val, ndisp, t in c.execute(q):
        short = name.split('(')[0][:60]
        if kernels and not any(k in short for k in kernels):
            continue
        a = agg.setdefault(short, {"n": ndisp, "t": t})
        a[names.get(pmc, str(pmc))] = val
    for kname, d in sorted(agg.items(), key=lambda kv: -kv[1]["t"]):
        print(f"\n{kname}  ({d['n']} dispatches, {d['t']/1e6:.2f} ms)")
        wc = d.get("SQ_WAVE_CYCLES")
        for cname in sorted(k for k in d if k not in ("n", "t")):
            v = d[cname]
            pct = f"  ({100*v/wc:.0f}% of wave cycles)" if wc and "WAIT" in cname or wc and "ACTIVE" in cname else ""
            pct = f"  ({100*v/wc:.0f}%)" if wc and cname != "SQ_WAVE_CYCLES" and ("WAIT" in cname or "ACTIVE" in cname or "MFMA" in cname) else ""
            print(f"  {cname:28s} {v:>16,}{pct}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2:])
