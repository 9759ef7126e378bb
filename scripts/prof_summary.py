#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into a kernel-stats table.

Usage: python scripts/prof_summary.py <results.db> [top_n]
Prints total-ms / calls / avg-us per kernel, descending total.
(rocprofv3 on this image emits SQLite .db instead of CSV stats.)"""

import sqlite3
import sys


def summarize(path, top=30):
    c = sqlite3.connect(path)
    tables = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith('rocpd_kernel_dispatch_'))
    ks = next(t for t in tables if t.startswith('rocpd_info_kernel_symbol_'))
    q = f"""
    SELECT ks.display_name, COUNT(*), AVG(k.end-k.start)/1e3,
           SUM(k.end-k.start)/1e6
    FROM {kd} k JOIN {ks} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 4 DESC LIMIT {top}
    """
    rows = list(c.execute(q))
    total = sum(r[3] for r in rows)
    print("| total ms | calls | avg us | % | kernel |")
    print("|---|---|---|---|---|")
    for name, n, avg, tot in rows:
        print(f"| {tot:.1f} | {n} | {avg:.1f} | {100*tot/total:.1f} "
              f"| `{name[:70]}` |")
    print(f"\n(sum of listed: {total:.1f} ms)")


if __name__ == '__main__':
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 30)
