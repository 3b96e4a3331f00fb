#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite database into a per-kernel hot list.

Usage: python tools/rocpd_stats.py <bench_results.db> [-n TOPK]
"""

import argparse
import re
import sqlite3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("-n", type=int, default=40)
    args = ap.parse_args()

    c = sqlite3.connect(args.db)
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))

    q = f"""
    SELECT ks.display_name AS name, COUNT(*) AS calls,
           SUM(k.end-k.start)/1e6 AS total_ms,
           AVG(k.end-k.start)/1e3 AS avg_us
    FROM {disp} k JOIN {sym} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY total_ms DESC
    """
    rows = c.execute(q).fetchall()
    tot = sum(r[2] for r in rows)
    print(f"{'kernel':<86} {'calls':>6} {'total_ms':>10} {'avg_us':>9} {'%':>6}")
    for name, calls, ms, avg in rows[:args.n]:
        n = re.sub(r"\(.*", "", name)[:84]
        print(f"{n:<86} {calls:>6} {ms:>10.2f} {avg:>9.1f} {100 * ms / tot:>6.2f}")
    print(f"# TOTAL gpu kernel time: {tot:.1f} ms; {len(rows)} distinct kernels")


if __name__ == "__main__":
    main()
