"""Summarize a rocpd sqlite db into a per-kernel time table (stdout)."""
import sqlite3, sys, re, glob
db = glob.glob(sys.argv[1])[0]
c = sqlite3.connect(db)
sfx = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")][0].replace('rocpd_kernel_dispatch_', '')
rows = c.execute(f"""
SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6, AVG(kd.end-kd.start)/1e3
FROM rocpd_kernel_dispatch_{sfx} kd
JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30""").fetchall()
tot = c.execute(f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{sfx}").fetchone()[0]
for name, n, ms, aus in rows:
    print(f'{ms:9.2f}ms {100*ms/tot:5.1f}% n={n:6d} avg={aus:7.1f}us  {re.sub(r"[(].*", "", name)[:70]}')
print(f'TOTAL {tot:.1f} ms')
