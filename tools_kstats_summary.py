"""Summarize a rocprofv3 kernel-trace results DB into per-kernel totals
(CSV: kernel, calls, total_ms, avg_ms, pct). Usage:
    python tools_kstats_summary.py <dir-with-*_results.db> <out.csv>
"""
import glob
import sqlite3
import sys
from collections import defaultdict

db_path = glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True)[0]
out = sys.argv[2]
db = sqlite3.connect(db_path)
cur = db.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
u = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0].split("rocpd_kernel_dispatch_")[1]

strings = dict(cur.execute(f"SELECT id, string FROM rocpd_string_{u}").fetchall())
ksym = {}
for kid, dn in cur.execute(f"SELECT id, display_name FROM rocpd_info_kernel_symbol_{u}").fetchall():
    ksym[kid] = strings.get(dn, dn if isinstance(dn, str) else str(dn))

cols = [r[1] for r in cur.execute(f"PRAGMA table_info(rocpd_kernel_dispatch_{u})")]
start_c = "start" if "start" in cols else "start_timestamp"
end_c = "end" if "end" in cols else "end_timestamp"

agg = defaultdict(lambda: [0, 0])
for kid, st, en in cur.execute(f"SELECT kernel_id, {start_c}, {end_c} FROM rocpd_kernel_dispatch_{u}"):
    name = ksym.get(kid, str(kid))
    agg[name[:100]][0] += 1
    agg[name[:100]][1] += en - st

total = sum(v[1] for v in agg.values())
rows = sorted(agg.items(), key=lambda kv: -kv[1][1])
with open(out, "w") as f:
    f.write("kernel,calls,total_ms,avg_ms,pct\n")
    for name, (n, ns) in rows:
        f.write(f'"{name}",{n},{ns / 1e6:.3f},{ns / 1e6 / n:.4f},{100 * ns / total:.2f}\n')
print("wrote", out, len(rows), "kernels,", f"{total / 1e6:.0f} ms total GPU time")
