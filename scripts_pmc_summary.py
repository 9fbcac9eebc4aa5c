"""Summarize a rocprofv3 --pmc rocpd db: per-kernel counter totals."""
import sqlite3
import sys

db = sys.argv[1]
con = sqlite3.connect(db)
cur = con.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]


def find(prefix):
    for t in tables:
        if t.startswith(prefix):
            return t
    raise RuntimeError(f"{prefix}* not in {tables}")


try:
    cr = find("rocpd_counter")  # counter records
except RuntimeError:
    print("tables:", tables)
    raise
# discover schema
cols = [r[1] for r in cur.execute(f"PRAGMA table_info({cr})")]
print("counter table:", cr, cols, file=sys.stderr)
disp = find("rocpd_kernel_dispatch")
sym = find("rocpd_info_kernel_symbol")
q = f"""
SELECT k.display_name, c.counter_id, SUM(c.value)
FROM {cr} c
JOIN {disp} d ON c.dispatch_id = d.dispatch_id
JOIN {sym} k ON d.kernel_id = k.id
GROUP BY k.display_name, c.counter_id
"""
try:
    rows = cur.execute(q).fetchall()
except Exception as e:
    print("query failed:", e)
    for t in tables:
        print(t, [r[1] for r in cur.execute(f"PRAGMA table_info({t})")])
    sys.exit(1)
# counter names
try:
    ci = find("rocpd_info_counter")
    names = dict(cur.execute(f"SELECT id, name FROM {ci}"))
except Exception:
    names = {}
agg = {}
for kname, cid, val in rows:
    agg.setdefault(kname.split("(")[0][:60], {})[names.get(cid, cid)] = val
for k, d in sorted(agg.items(), key=lambda kv: -max(kv[1].values())):
    print(k)
    for n, v in sorted(d.items(), key=str):
        print(f"    {n}: {v:,.0f}")
