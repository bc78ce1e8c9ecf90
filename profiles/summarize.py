#!/usr/bin/env python3
"""Summarize a rocprofv3 result DB (--kernel-trace [-o name]) into the
per-kernel table committed under profiles/.

Usage: python profiles/summarize.py <results.db> [label]
"""
import sqlite3
import sys


def summarize(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    sfx = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'"
    )][0].replace('rocpd_kernel_dispatch_', '')
    q = f"""SELECT ks.display_name, COUNT(*) n,
                   SUM(kd.end-kd.start)/1e6 ms, AVG(kd.end-kd.start)/1e3 avg
            FROM rocpd_kernel_dispatch_{sfx} kd
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY ms DESC"""
    lines = []
    tot = 0.0
    for name, n, ms, avg in cur.execute(q).fetchall():
        tot += ms
        lines.append(f"{name.split('(')[0][:52]:52s} n={n:6d} "
                     f"total={ms:9.3f}ms avg={avg:9.3f}us")
    rows = cur.execute(
        f"SELECT start,end FROM rocpd_kernel_dispatch_{sfx} ORDER BY start"
    ).fetchall()
    span = (rows[-1][1] - rows[0][0]) / 1e6
    gaps = sum(max(0, rows[i + 1][0] - rows[i][1])
               for i in range(len(rows) - 1)) / 1e6
    lines.append(f"TOTAL kernel time {tot:.2f} ms; dispatch span {span:.2f} ms; "
                 f"inter-dispatch gaps {gaps:.2f} ms; {len(rows)} dispatches")
    return "\n".join(lines)


if __name__ == "__main__":
    print(summarize(sys.argv[1]))
