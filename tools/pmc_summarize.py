#!/usr/bin/env python3
"""Summarize a rocprofv3 sqlite output dir: per-kernel totals of time and PMC
counters. Usage: python tools/pmc_summarize.py <dir-with-db> """
import glob
import sqlite3
import sys
from collections import defaultdict

d = sys.argv[1]
db = sorted(glob.glob(d + "/*.db"))[0]
con = sqlite3.connect(db)
cur = con.cursor()
names = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = [n for n in names if n.startswith("rocpd_kernel_dispatch")][0].replace(
    "rocpd_kernel_dispatch_", "")

pmc_cols = [r[1] for r in cur.execute(f"PRAGMA table_info(rocpd_pmc_event_{sfx})")]
print("# pmc_event cols:", pmc_cols, file=sys.stderr)
pmc_info = {}
try:
    for r in cur.execute(f"SELECT id, name FROM rocpd_info_pmc_{sfx}"):
        pmc_info[r[0]] = r[1]
except Exception as e:
    print("# pmc_info err:", e, file=sys.stderr)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info(rocpd_info_pmc_{sfx})")]
    print("# pmc_info cols:", cols, file=sys.stderr)
    for r in cur.execute(f"SELECT * FROM rocpd_info_pmc_{sfx}"):
        print("#", r, file=sys.stderr)

# kernel time
time_rows = cur.execute(f"""
SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6
FROM rocpd_kernel_dispatch_{sfx} k
JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id
GROUP BY 1 ORDER BY 3 DESC LIMIT 25""").fetchall()
for name, n, ms in time_rows:
    print(f"TIME\t{ms:.3f}\t{n}\t{name[:90]}")

# counters joined via the dispatch/event linkage (schema varies; try both)
for join in (
    f"JOIN rocpd_kernel_dispatch_{sfx} k ON p.event_id=k.event_id",
    f"JOIN rocpd_kernel_dispatch_{sfx} k ON p.event_id=k.id",
):
    try:
        rows = cur.execute(f"""
        SELECT ks.display_name, p.pmc_id, SUM(p.value)
        FROM rocpd_pmc_event_{sfx} p
        {join}
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id
        GROUP BY 1,2""").fetchall()
        if rows:
            agg = defaultdict(dict)
            for name, pid, val in rows:
                agg[name[:90]][pmc_info.get(pid, str(pid))] = val
            for name, dd in sorted(agg.items()):
                print(f"PMC\t{name}")
                for c, v in sorted(dd.items()):
                    print(f"  {c}\t{v:,.0f}")
            break
    except Exception as e:
        print("# join failed:", e, file=sys.stderr)
