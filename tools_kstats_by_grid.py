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


def modal_split(db_cur, u, ksym, name_filter, mod, labels):
    """Bucket a kernel's dispatches by (order index % mod) — the per-layer
    dispatch cycle is deterministic, so this splits shared Tensile kernels
    by GEMM shape."""
    rows = []
    for kid, st, en in db_cur.execute(
        f"SELECT kernel_id, {start_c}, {end_c} FROM rocpd_kernel_dispatch_{u} ORDER BY {start_c}"
    ):
        if name_filter in ksym.get(kid, ""):
            rows.append(en - st)
    # drop the trailing odd calls (lm_head has its own grid bucket anyway)
    n = len(rows) - (len(rows) % mod)
    agg = defaultdict(lambda: [0, 0])
    for i, d in enumerate(rows[:n]):
        agg[i % mod][0] += 1
        agg[i % mod][1] += d
    for m in range(mod):
        c, t = agg[m]
        lab = labels[m] if m < len(labels) else str(m)
        print(f"  pos {m} ({lab:8s}): calls={c:4d} avg={t/1e6/max(c,1):7.3f}ms")


if len(sys.argv) > 3 and sys.argv[3] == "mod4":
    print("MT256x256x32 fwd calls by position in the layer cycle:")
    modal_split(cur, u, ksym, "MT256x256x32", 4, ["c_attn", "c_proj", "c_fc", "mlp_proj"])
