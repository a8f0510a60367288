"""Summarize a rocprofv3 --pmc SQ_INSTS_MFMA sqlite db: total counter
value per kernel symbol.  Usage: python pmc_dump.py <db-glob>"""
import glob
import sqlite3
import sys

dbs = sorted(glob.glob(sys.argv[1]))
print("dbs:", dbs)
db = sqlite3.connect(dbs[0])
tables = [r[0] for r in db.execute("SELECT name FROM sqlite_master WHERE type='table'")]
ct = [t for t in tables if "counter" in t.lower()]
print("counter tables:", ct)
for t in ct:
    cols = [r[1] for r in db.execute(f"PRAGMA table_info({t})")]
    print(t, cols)
q = """
SELECT s.display_name, SUM(c.value), COUNT(*)
FROM rocpd_counter c
JOIN rocpd_kernel_dispatch k ON c.dispatch_id = k.dispatch_id
JOIN rocpd_info_kernel_symbol s ON k.kernel_id = s.id
GROUP BY s.display_name ORDER BY 2 DESC LIMIT 25
"""
try:
    rows = db.execute(q).fetchall()
except Exception as e:
    print("join failed:", e)
    rows = []
for n, v, c in rows:
    print(f"{int(v):>15,d} n={c:5d}  {n[:75]}")
