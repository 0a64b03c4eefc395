#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db: per-kernel time (+ optional PMC)."""
import glob
import sqlite3
import sys

db_path = glob.glob(sys.argv[1])[0]
db = sqlite3.connect(db_path)
tables = [r[0] for r in db.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
sfx = next(t for t in tables if t.startswith("rocpd_kernel_dispatch_")
           ).split("rocpd_kernel_dispatch_")[1]
q = f"""
SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
       AVG(k.end-k.start)/1e3
FROM rocpd_kernel_dispatch_{sfx} k
JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 15
"""
print(f"{'total_ms':>9} {'calls':>6} {'avg_us':>9}  kernel")
for name, calls, tot, avg in db.execute(q):
    print(f"{tot:9.2f} {calls:6d} {avg:9.1f}  {name[:90]}")
pmc = f"rocpd_pmc_event_{sfx}"
if pmc in tables:
    cols = [r[1] for r in db.execute(f"PRAGMA table_info({pmc})")]
    print("\n# PMC table columns:", cols)
    q2 = f"""
    SELECT ks.display_name, i.name, SUM(p.value), COUNT(*)
    FROM {pmc} p
    JOIN rocpd_info_pmc_{sfx} i ON p.pmc_id = i.id
    JOIN rocpd_kernel_dispatch_{sfx} k ON k.event_id = p.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name, i.name
    HAVING SUM(p.value) > 0 ORDER BY 1, 2
    """
    try:
        cur = None
        for name, ctr, tot, n in db.execute(q2):
            if name != cur:
                print(f"\n== {name[:90]} ({n} samples)")
                cur = name
            print(f"   {ctr:24s} {tot:,.0f}")
    except sqlite3.OperationalError as ex:
        print("PMC join failed:", ex)
        for t in tables:
            if "pmc" in t:
                print(t, [r[1] for r in db.execute(f"PRAGMA table_info({t})")])
