"""Summarize a rocprofv3 result (SQLite .db from --stats/--kernel-trace, or
a kernel_stats.csv) into a per-kernel table: calls, total ms, %, avg us.

Usage: python tools/prof_summary.py <results.db|kernel_stats.csv> [top_n]
"""
from __future__ import annotations

import csv
import sys


def from_db(path):
    import sqlite3

    c = sqlite3.connect(path)
    tables = [r[0] for r in c.execute("select name from sqlite_master where type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    q = f"""
      select s.display_name, count(*), sum(d.end - d.start), avg(d.end - d.start)
      from {disp} d join {sym} s on d.kernel_id = s.id
      group by s.display_name order by sum(d.end - d.start) desc
    """
    return [(name, n, tot, avg) for name, n, tot, avg in c.execute(q)]


def from_csv(path):
    rows = list(csv.DictReader(open(path)))
    out = []
    for r in rows:
        out.append((r["Name"], int(r["Calls"]), float(r["TotalDurationNs"]),
                    float(r["AverageNs"])))
    out.sort(key=lambda x: -x[2])
    return out


def main():
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    rows = from_db(path) if path.endswith(".db") else from_csv(path)
    total = sum(r[2] for r in rows)
    print(f"{'kernel':<72} {'calls':>6} {'total_ms':>10} {'%':>6} {'avg_us':>9}")
    for name, n, tot, avg in rows[:top]:
        short = name if len(name) <= 70 else name[:67] + "..."
        print(f"{short:<72} {n:>6} {tot/1e6:>10.2f} {100*tot/total:>6.2f} {avg/1e3:>9.1f}")
    print(f"{'TOTAL':<72} {sum(r[1] for r in rows):>6} {total/1e6:>10.2f}")


if __name__ == "__main__":
    main()
