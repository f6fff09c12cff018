"""Summarize rocprofv3 --pmc rocpd SQLite DBs: per-kernel average counter value.

Usage: python tools/pmc_summarize.py <dir-with-runc/*.db> [...]
Prints one line per kernel (top 6 by total) per DB.
"""
import glob
import re
import sqlite3
import sys

for root in sys.argv[1:]:
    dbs = sorted(glob.glob(f"{root}/runc/*.db")) or sorted(glob.glob(f"{root}/*.db"))
    for db in dbs:
        con = sqlite3.connect(db)
        names = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'")]

        def tab(sub):
            m = [t for t in names if sub in t]
            return m[0] if m else None

        ev, kd, ks, info = tab("pmc_event"), tab("kernel_dispatch"), tab("info_kernel_symbol"), tab("info_pmc")
        if not (ev and kd and ks):
            continue
        counter = "?"
        if info:
            row = con.execute(f"SELECT name FROM {info} LIMIT 1").fetchone()
            if row:
                counter = row[0]
        print(f"== {root} counter={counter} ==")
        q = f"""SELECT k.display_name, COUNT(*), AVG(e.value), SUM(e.value)
                FROM {ev} e JOIN {kd} d ON e.event_id=d.id JOIN {ks} k ON d.kernel_id=k.id
                GROUP BY 1 ORDER BY 4 DESC LIMIT 6"""
        for name, n, avg, tot in con.execute(q):
            nm = re.sub(r"\(anonymous namespace\)::", "", str(name))[:56]
            print(f"  {nm:58s} n={n:3d} avg={avg:16,.0f} tot={tot:18,.0f}")
        con.close()
