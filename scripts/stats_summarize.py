#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite DB into a per-kernel time table.

Usage: stats_summarize.py <db_path> [top_n]
"""

import sqlite3
import sys


def main():
    db, top_n = sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 40
    con = sqlite3.connect(db)
    rows = con.execute(
        "select name, count(*), sum(end-start) from kernels group by name"
    ).fetchall()
    total = sum(r[2] for r in rows)
    n_disp = sum(r[1] for r in rows)
    print(f"# {n_disp} dispatches, GPU busy {total/1e6:.1f} ms")
    for name, cnt, dur in sorted(rows, key=lambda r: -r[2])[:top_n]:
        print(f"{dur/1e6:9.2f} ms {100*dur/total:5.1f}% {cnt:6d}  {name[:96]}")


if __name__ == '__main__':
    main()
