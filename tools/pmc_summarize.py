"""Summarize a rocprofv3 PMC db: per-kernel counter totals (run on GPU box)."""
import glob
import sqlite3
import sys

db = sorted(glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True))[0]
con = sqlite3.connect(db)
cur = con.cursor()
names = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = [n for n in names if n.startswith("rocpd_kernel_dispatch")][0].split("rocpd_kernel_dispatch_")[1]
# pmc_event table links events to dispatches
q = f"""
SELECT s.display_name, i.name, SUM(e.value), COUNT(DISTINCT e.event_id)
FROM rocpd_pmc_event_{sfx} e
JOIN rocpd_info_pmc_{sfx} i ON e.pmc_id = i.id
JOIN rocpd_kernel_dispatch_{sfx} k ON e.event_id = k.event_id
JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id = s.id
GROUP BY s.display_name, i.name
"""
try:
    rows = cur.execute(q).fetchall()
except sqlite3.OperationalError:
    rows = []
agg = {}
for name, ctr, val, cnt in rows:
    agg.setdefault(name[:56], {})[ctr] = (val, cnt)
# also wall time
q2 = f"""SELECT s.display_name, SUM(k.end-k.start)/1000.0, COUNT(*)
FROM rocpd_kernel_dispatch_{sfx} k JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id=s.id
GROUP BY s.display_name"""
wall = {r[0][:56]: (r[1], r[2]) for r in cur.execute(q2).fetchall()}
print(f"{'kernel':<56} {'calls':>5} {'us_tot':>9} {'avg_us':>8} | counters per call")
keys = agg.keys() if agg else wall.keys()
for name in sorted(keys, key=lambda k: -wall.get(k, (0, 1))[0]):
    w, c = wall.get(name, (0, 1))
    line = f"{name:<56} {c:>5} {w:>9.0f} {w/max(1,c):>8.2f} | "
    for ctr, (val, cnt) in sorted(agg.get(name, {}).items()):
        line += f"{ctr.replace('SQ_','')}={val/max(1,c):.3g} "
    print(line)
