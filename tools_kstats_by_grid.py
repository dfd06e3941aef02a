"""Per-(kernel, grid) timing from a rocprofv3 kernel-trace DB — splits the
shared Tensile GEMM kernels by launch geometry so each GEMM shape's
in-step throughput is visible. Usage: tools_kstats_by_grid.py <dir> [filter]"""
import glob
import sqlite3
import sys
from collections import defaultdict

db_path = glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True)[0]
filt = sys.argv[2] if len(sys.argv) > 2 else ""
db = sqlite3.connect(db_path)
cur = db.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
u = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0].split("rocpd_kernel_dispatch_")[1]
strings = dict(cur.execute(f"SELECT id, string FROM rocpd_string_{u}").fetchall())
ksym = {}
for kid, dn in cur.execute(f"SELECT id, display_name FROM rocpd_info_kernel_symbol_{u}").fetchall():
    ksym[kid] = strings.get(dn, dn if isinstance(dn, str) else str(dn))
cols = [r[1] for r in cur.execute(f"PRAGMA table_info(rocpd_kernel_dispatch_{u})")]
print("# columns:", cols, file=sys.stderr)
start_c = "start" if "start" in cols else "start_timestamp"
end_c = "end" if "end" in cols else "end_timestamp"
gx = [c for c in cols if "grid" in c.lower()]
sel = f"SELECT kernel_id, {start_c}, {end_c}, {', '.join(gx[:3])} FROM rocpd_kernel_dispatch_{u}"
agg = defaultdict(lambda: [0, 0])
for row in cur.execute(sel):
    kid, st, en = row[0], row[1], row[2]
    grid = tuple(row[3:])
    name = ksym.get(kid, str(kid))
    if filt and filt not in name:
        continue
    key = (name[:60], grid)
    agg[key][0] += 1
    agg[key][1] += en - st
for (name, grid), (cnt, tot) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
    print(f"{name[:58]:60s} grid={str(grid):24s} calls={cnt:4d} total={tot/1e6:9.1f}ms avg={tot/1e6/cnt:7.3f}ms")
