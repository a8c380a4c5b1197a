#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db (kernel-trace) into a per-kernel table:
count, total ms, avg ms.  Usage: python tools/prof_summary.py <results.db> [title]"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    title = sys.argv[2] if len(sys.argv) > 2 else path
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6, AVG(kd.end-kd.start)/1e6
    FROM {kd} kd JOIN {ks} ks ON kd.kernel_id = ks.id
    GROUP BY 1 ORDER BY 3 DESC
    """
    print(f"rocprofv3 kernel-trace stats — {title}")
    print(f"{'kernel':32s} {'n':>6s} {'total_ms':>10s} {'avg_ms':>9s}")
    for name, n, tot, avg in cur.execute(q):
        print(f"{name[:32]:32s} {n:6d} {tot:10.3f} {avg:9.4f}")


if __name__ == "__main__":
    main()
