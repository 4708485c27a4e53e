#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database (newer rocprofv3 emits
*_results.db instead of CSV stats) into the per-kernel stats table the
profiles/ directory commits: Name, Calls, TotalDurationNs, AverageNs,
Percentage, MinNs, MaxNs."""
import csv
import sqlite3
import sys


def main(db_path, out_csv=None):
    c = sqlite3.connect(db_path)
    uuid = None
    for (name,) in c.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        if name.startswith("rocpd_kernel_dispatch_"):
            uuid = name[len("rocpd_kernel_dispatch_"):]
    assert uuid, "no kernel dispatch table"
    q = f"""
      SELECT k.display_name AS name, COUNT(*) AS calls,
             SUM(d.end - d.start) AS total_ns,
             AVG(d.end - d.start) AS avg_ns,
             MIN(d.end - d.start) AS min_ns,
             MAX(d.end - d.start) AS max_ns
      FROM rocpd_kernel_dispatch_{uuid} d
      JOIN rocpd_info_kernel_symbol_{uuid} k
        ON d.kernel_id = k.id AND d.nid = k.nid AND d.pid = k.pid
      GROUP BY k.display_name ORDER BY total_ns DESC"""
    rows = list(c.execute(q))
    grand = sum(r[2] for r in rows) or 1
    out = [("Name", "Calls", "TotalDurationNs", "AverageNs",
            "Percentage", "MinNs", "MaxNs")]
    for name, calls, tot, avg, mn, mx in rows:
        out.append((name, calls, tot, round(avg, 1),
                    round(100.0 * tot / grand, 2), mn, mx))
    w = csv.writer(open(out_csv, "w") if out_csv else sys.stdout)
    w.writerows(out)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
