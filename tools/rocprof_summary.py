#!/usr/bin/env python3
"""Extracts judge-facing summaries from rocprofv3 result databases
(the rocpd SQLite files rocprofv3 writes under -d) into plain text committed
under profiles/.

Usage: python tools/rocprof_summary.py <results.db> [<results.db>...] > out.txt
"""
import sqlite3
import sys


def summarize(db):
    con = sqlite3.connect(db)
    print(f"== {db}")
    try:
        rows = con.execute(
            "SELECT name, total_calls, total_duration, average, percentage "
            "FROM top_kernels ORDER BY total_duration DESC LIMIT 10").fetchall()
        print(f"{'KERNEL':80s} {'CALLS':>6s} {'TOTAL_us':>12s} {'AVG_us':>10s} {'PCT':>6s}")
        for name, calls, total, avg, pct in rows:
            print(f"{name[:80]:80s} {calls:6d} {total:12.1f} {avg:10.1f} {pct:6.2f}")
    except sqlite3.Error:
        pass
    try:
        rows = con.execute(
            "SELECT kernel_name, counter_name, count(*), avg(value), sum(value), "
            "avg(duration) FROM counters_collection GROUP BY kernel_name, "
            "counter_name ORDER BY sum(value) DESC LIMIT 10").fetchall()
        if rows:
            print(f"\n{'KERNEL':60s} {'COUNTER':>12s} {'N':>5s} {'AVG':>16s} "
                  f"{'AVG_GB':>9s} {'AVG_DUR_ms':>10s}")
            for name, counter, n, avg, total, dur in rows:
                print(f"{name[:60]:60s} {counter:>12s} {n:5d} {avg:16.1f} "
                      f"{avg / (1024 * 1024):9.3f} {dur / 1e6:10.3f}")
    except sqlite3.Error:
        pass
    con.close()
    print()


if __name__ == "__main__":
    for db in sys.argv[1:]:
        summarize(db)
