#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db: per-kernel total time, calls, avg.

Usage: python scripts/rocpd_stats.py <results.db> [top_n]
"""
import sqlite3
import sys


def main():
    db, top = sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 30
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    q = f"""
        SELECT s.display_name, COUNT(*) AS calls,
               SUM(d.end - d.start) / 1e6 AS total_ms,
               AVG(d.end - d.start) / 1e3 AS avg_us
        FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY total_ms DESC LIMIT {top}
    """
    rows = list(cur.execute(q))
    tot = cur.execute(f"SELECT SUM(end - start) / 1e6 FROM {kd}").fetchone()[0]
    span = cur.execute(f"SELECT (MAX(end) - MIN(start)) / 1e6 FROM {kd}").fetchone()[0]
    ncalls = cur.execute(f"SELECT COUNT(*) FROM {kd}").fetchone()[0]
    print(f"kernel-busy total: {tot:.1f} ms over wall span {span:.1f} ms "
          f"({100 * tot / span:.1f}% busy), {ncalls} dispatches")
    print(f"{'total_ms':>9} {'calls':>7} {'avg_us':>8}  name")
    for name, calls, ms, avg in rows:
        print(f"{ms:9.2f} {calls:7d} {avg:8.1f}  {name[:110]}")


if __name__ == "__main__":
    main()
