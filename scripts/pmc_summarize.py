#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd database's PMC counters per kernel (run on
the GPU box; keeps only a small text summary)."""
import sqlite3
import sys
from collections import defaultdict

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
try:
    rows = list(cur.execute("""
        SELECT k.name, p.name, SUM(e.value), COUNT(*)
        FROM pmc_events e
        JOIN kernels k ON e.dispatch_id = k.dispatch_id
        JOIN pmc_info p ON e.pmc_id = p.id
        GROUP BY k.name, p.name"""))
except Exception as exc:
    # schema fallback: dump table names
    print('query failed:', exc)
    for r in cur.execute("SELECT name FROM sqlite_master WHERE type IN ('table','view')"):
        print(r[0])
    sys.exit(0)
per = defaultdict(dict)
for kname, cname, val, cnt in rows:
    per[kname][cname] = (val, cnt)
for kname, counters in sorted(per.items()):
    short = kname.split('(')[0][:70]
    print(f'== {short}')
    for cname, (val, cnt) in sorted(counters.items()):
        print(f'   {cname:28s} sum={val:.3e} dispatches={cnt}')
