#!/usr/bin/env python3
"""Kernel-stats CSV (name, total_calls, total_duration, average, percentage)
from a rocprofv3 rocpd SQLite database (ROCm 7.2 default output format).

Usage: python tools/rocpd_stats.py <results.db> [out.csv]
"""

import csv
import sqlite3
import sys


def kernel_stats(db_path):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start)
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """).fetchall()
    total = sum(r[2] for r in rows) or 1
    out = []
    for name, calls, dur_ns in rows:
        us = dur_ns / 1e3
        out.append({
            "name": name,
            "total_calls": calls,
            "total_duration": round(us, 3),       # microseconds
            "average": round(us / calls, 3),
            "percentage": 100.0 * dur_ns / total,
        })
    return out


def main():
    db = sys.argv[1]
    rows = kernel_stats(db)
    w = csv.DictWriter(sys.stdout if len(sys.argv) < 3
                       else open(sys.argv[2], "w", newline=""),
                       fieldnames=list(rows[0]))
    w.writeheader()
    w.writerows(rows)


if __name__ == "__main__":
    main()
