#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd database's PMC counters per kernel (runs on
the GPU box; emits a small text summary instead of shipping the db)."""
import sqlite3
import sys
from collections import defaultdict

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()


def cols(table):
    try:
        return [r[1] for r in cur.execute(f'PRAGMA table_info({table})')]
    except sqlite3.Error:
        return []


candidates = ['counters_collection', 'pmc_events', 'rocpd_pmc_event']
for t in candidates:
    c = cols(t)
    if c:
        print(f'# table {t}: {c}', file=sys.stderr)

# preferred: the convenience view joining everything
cc = cols('counters_collection')
if cc:
    name_col = 'kernel_name' if 'kernel_name' in cc else None
    ctr_col = 'counter_name' if 'counter_name' in cc else None
    val_col = 'value' if 'value' in cc else None
    if name_col and ctr_col and val_col:
        per = defaultdict(lambda: defaultdict(lambda: [0.0, 0]))
        for kname, cname, val in cur.execute(
                f'SELECT {name_col}, {ctr_col}, {val_col} FROM counters_collection'):
            a = per[kname][cname]
            a[0] += val or 0.0
            a[1] += 1
        for kname in sorted(per):
            print(f'== {str(kname).split("(")[0][:70]}')
            for cname, (val, cnt) in sorted(per[kname].items()):
                print(f'   {cname:28s} sum={val:.4e} n={cnt}')
        sys.exit(0)

print('counters_collection view not usable; columns dumped to stderr')
