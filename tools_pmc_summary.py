"""Summarize a rocprofv3 PMC results DB into per-kernel mean counters (CSV).
Pure-python aggregation (schema ids are composite across rocprofv3 builds)."""
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

# counter id -> name (info_pmc.name may be a string id or raw text)
pmc_names = {}
for row in cur.execute(f"SELECT id, name, symbol FROM rocpd_info_pmc_{u}").fetchall():
    pid, name, symbol = row
    nm = strings.get(name, name if isinstance(name, str) else None) or strings.get(symbol, symbol)
    pmc_names[pid] = str(nm)
print("info_pmc rows:", len(pmc_names), "sample ids:", list(pmc_names.items())[:3])

# kernel symbol id -> display name
ksym = {}
for kid, dn in cur.execute(f"SELECT id, display_name FROM rocpd_info_kernel_symbol_{u}").fetchall():
    ksym[kid] = strings.get(dn, dn if isinstance(dn, str) else str(dn))

# dispatch event_id -> kernel name
ev2k = {}
for eid, kid in cur.execute(f"SELECT event_id, kernel_id FROM rocpd_kernel_dispatch_{u}").fetchall():
    ev2k[eid] = ksym.get(kid, str(kid))

agg = defaultdict(lambda: [0, 0.0])
unmatched_pmc = set()
for eid, pid, val in cur.execute(f"SELECT event_id, pmc_id, value FROM rocpd_pmc_event_{u}").fetchall():
    kname = ev2k.get(eid)
    if kname is None:
        continue
    cname = pmc_names.get(pid)
    if cname is None:
        # composite id: try low 32 bits
        cname = pmc_names.get(pid & 0xFFFFFFFF)
        if cname is None:
            unmatched_pmc.add(pid)
            continue
    key = (kname[:90], cname)
    agg[key][0] += 1
    agg[key][1] += val

if unmatched_pmc:
    print("unmatched pmc ids:", list(unmatched_pmc)[:5])
with open(out, "w") as f:
    f.write("kernel,counter,samples,avg,total\n")
    for (kn, cn), (n, tot) in sorted(agg.items()):
        f.write(f'"{kn}",{cn},{n},{tot/n:.1f},{tot:.0f}\n')
print("wrote", out, len(agg))
