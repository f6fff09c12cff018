"""Summarize rocprofv3 rocpd SQLite traces into a compact kernel table."""
import glob, re, sqlite3, sys

root = sys.argv[1]
for cfg_dir in sorted(glob.glob(f"{root}/c*/")):
    dbs = glob.glob(cfg_dir + "runc/*.db")
    if not dbs:
        continue
    con = sqlite3.connect(dbs[0])
    names = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    def tab(sub):
        return [t for t in names if sub in t][0]
    kd, ks = tab("kernel_dispatch"), tab("info_kernel_symbol")
    print(f"== {cfg_dir} ==")
    q = f"""SELECT k.display_name, COUNT(*), AVG(d.end-d.start), SUM(d.end-d.start),
            MAX(d.grid_size_x), MAX(d.grid_size_y), MAX(d.workgroup_size_x), MAX(d.group_segment_size)
            FROM {kd} d JOIN {ks} k ON d.kernel_id=k.id GROUP BY 1 ORDER BY 4 DESC LIMIT 10"""
    for name, n, avg, tot, gx, gy, wx, lds in con.execute(q):
        nm = re.sub(r"\(anonymous namespace\)::", "", str(name))[:64]
        print(f"  {nm:66s} n={n:3d} avg={avg/1e6:8.3f}ms tot={tot/1e6:8.2f}ms grid={gx}x{gy} wg={wx} lds={lds}")
    con.close()
