"""Summarize a rocprofv3 PC-sampling db: histogram of samples per kernel and
per instruction offset within the hottest kernel. Run on the GPU box."""
import glob
import sqlite3
import sys

dbs = sorted(glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True))
if not dbs:
    print("no db under", sys.argv[1])
    sys.exit(1)
con = sqlite3.connect(dbs[-1])
cur = con.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
pct = [t for t in tables if "pc_sampl" in t.lower() and "info" not in t.lower()]
print("tables:", [t for t in tables if "pc" in t.lower() or "code" in t.lower()])
if not pct:
    sys.exit("no pc sampling table")
t = pct[0]
cols = [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
print(t, "columns:", cols)
# find kernel-name-ish and offset-ish columns generically
rows = cur.execute(f"SELECT * FROM {t} LIMIT 3").fetchall()
for r in rows:
    print(r)
# common schema: code_object_id + code_object_offset; join kernel symbol if possible
off_col = next((c for c in cols if "offset" in c), None)
sym = [tt for tt in tables if "kernel_symbol" in tt]
name_expr = None
if "dispatch_id" in cols and sym:
    sfx = sym[0].split("rocpd_info_kernel_symbol_")[-1]
    disp = [tt for tt in tables if tt.startswith("rocpd_kernel_dispatch")]
    if disp:
        q = f"""SELECT s.display_name, COUNT(*) FROM {t} p
                JOIN {disp[0]} k ON p.dispatch_id = k.dispatch_id
                JOIN {sym[0]} s ON k.kernel_id = s.id
                GROUP BY s.display_name ORDER BY 2 DESC"""
        try:
            for name, n in cur.execute(q).fetchall():
                print(f"{n:>8}  {name[:70]}")
            q2 = f"""SELECT p.{off_col}, COUNT(*) FROM {t} p
                JOIN {disp[0]} k ON p.dispatch_id = k.dispatch_id
                JOIN {sym[0]} s ON k.kernel_id = s.id
                WHERE s.display_name LIKE '%dkv%'
                GROUP BY p.{off_col} ORDER BY 2 DESC LIMIT 60"""
            print("top offsets in dkv kernel:")
            for off, n in cur.execute(q2).fetchall():
                print(f"{n:>8}  0x{off:x}" if isinstance(off, int) else f"{n:>8}  {off}")
        except sqlite3.OperationalError as e:
            print("join failed:", e)
elif off_col:
    q = f"SELECT {off_col}, COUNT(*) FROM {t} GROUP BY 1 ORDER BY 2 DESC LIMIT 60"
    for off, n in cur.execute(q).fetchall():
        print(f"{n:>8}  {off}")
