"""Summarize a rocprofv3 PMC results DB into per-kernel mean counters (CSV)."""
import glob
import sqlite3
import sys

db_path = glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True)[0]
out = sys.argv[2]
db = sqlite3.connect(db_path)
cur = db.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
u = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0].split("rocpd_kernel_dispatch_")[1]

def cols(t):
    return [r[1] for r in cur.execute(f"PRAGMA table_info({t}_{u})")]

print("pmc_event cols:", cols("rocpd_pmc_event"))
print("info_pmc cols:", cols("rocpd_info_pmc"))
print("counts:", cur.execute(f"SELECT COUNT(*) FROM rocpd_pmc_event_{u}").fetchone(),
      cur.execute(f"SELECT COUNT(*) FROM rocpd_kernel_dispatch_{u}").fetchone())
print("sample pmc rows:", cur.execute(f"SELECT * FROM rocpd_pmc_event_{u} LIMIT 3").fetchall())
print("sample dispatch id/event:", cur.execute(f"SELECT id,event_id,kernel_id FROM rocpd_kernel_dispatch_{u} LIMIT 3").fetchall())

pe = cols("rocpd_pmc_event")
join_col = "event_id" if "event_id" in pe else pe[0]
# try join on event_id first, then dispatch_id
for jc, kd in [("event_id", "event_id"), ("dispatch_id", "dispatch_id"), ("event_id", "id")]:
    if jc not in pe:
        continue
    try:
        rows = cur.execute(f"""
            SELECT ks.display_name, s.string, COUNT(*), AVG(p.value), SUM(p.value)
            FROM rocpd_pmc_event_{u} p
            JOIN rocpd_kernel_dispatch_{u} k ON p.{jc} = k.{kd}
            JOIN rocpd_info_kernel_symbol_{u} ks ON k.kernel_id = ks.id
            JOIN rocpd_info_pmc_{u} pi ON p.pmc_id = pi.id
            JOIN rocpd_string_{u} s ON pi.name = s.id
            GROUP BY ks.display_name, s.string""").fetchall()
    except Exception as e:
        print("join", jc, kd, "failed:", e)
        continue
    if rows:
        with open(out, "w") as f:
            f.write("kernel,counter,dispatches,avg,total\n")
            for r in rows:
                f.write(f'"{r[0][:90]}",{r[1]},{r[2]},{r[3]:.1f},{r[4]:.0f}\n')
        print("wrote", out, len(rows), "via", jc, kd)
        break
else:
    print("no join worked")
