#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite database: total time and calls per
kernel, descending.  Usage: python tools/analyze_prof.py <db> [top_n]"""

import sqlite3
import sys


def main():
    path = sys.argv[1]
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    assert tabs, "no kernel dispatch table"
    sfx = tabs[0][len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
               MIN(kd.start), MAX(kd.end)
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}""").fetchall()
    total = cur.execute(f"SELECT SUM(end-start)/1e6 FROM "
                        f"rocpd_kernel_dispatch_{sfx}").fetchone()[0]
    print(f"{'ms':>10} {'calls':>7}  kernel")
    for name, cnt, ms, *_ in rows:
        print(f"{ms:10.1f} {cnt:7d}  {name[:110]}")
    print(f"TOTAL gpu kernel ms: {total:.1f}")


if __name__ == "__main__":
    main()
