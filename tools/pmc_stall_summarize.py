"""Summarize a rocprofv3 --pmc run with multiple counters: per-kernel average
of each counter. Usage: python tools/pmc_stall_summarize.py <dir> [name-filter]"""
import glob
import re
import sqlite3
import sys

root = sys.argv[1]
like = sys.argv[2] if len(sys.argv) > 2 else ""
dbs = sorted(glob.glob(f"{root}/runc/*.db")) or sorted(glob.glob(f"{root}/*.db"))
for db in dbs:
    con = sqlite3.connect(db)
    names = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(sub):
        m = [t for t in names if sub in t]
        return m[0] if m else None

    ev, kd, ks, info = tab("pmc_event"), tab("kernel_dispatch"), tab("info_kernel_symbol"), tab("info_pmc")
    if not (ev and kd and ks and info):
        continue
    q = f"""SELECT k.display_name, i.name, COUNT(*), AVG(e.value)
            FROM {ev} e JOIN {kd} d ON e.event_id=d.id JOIN {ks} k ON d.kernel_id=k.id
            JOIN {info} i ON e.pmc_id=i.id
            GROUP BY 1,2 ORDER BY 1,2"""
    for nm, c, n, avg in con.execute(q):
        s = re.sub(r"\(anonymous namespace\)::", "", str(nm))
        if like and like not in s:
            continue
        print(f"{re.sub(r'<.*', '', s)[:30]:32s} {c:26s} n={n:3d} avg={avg:20,.0f}")
    con.close()
