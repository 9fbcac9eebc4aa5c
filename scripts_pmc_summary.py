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


cr = find("rocpd_pmc_event")
disp = find("rocpd_kernel_dispatch")
sym = find("rocpd_info_kernel_symbol")
ci = find("rocpd_info_pmc")
for t in (cr, disp, ci):
    print(t.split("_0000")[0], [r[1] for r in cur.execute(f"PRAGMA table_info({t})")],
          file=sys.stderr)
cr_cols = [r[1] for r in cur.execute(f"PRAGMA table_info({cr})")]
ci_cols = [r[1] for r in cur.execute(f"PRAGMA table_info({ci})")]
# column-name guesses across rocpd versions
val_col = "value" if "value" in cr_cols else cr_cols[-1]
pmc_key = [c for c in cr_cols if "pmc" in c and c != val_col][0]
evt_key = [c for c in cr_cols if "event" in c or "dispatch" in c][0]
name_col = "name" if "name" in ci_cols else ci_cols[1]
ci_id = "id" if "id" in ci_cols else ci_cols[0]
disp_cols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]
d_evt = [c for c in disp_cols if "event" in c or c == "dispatch_id" or c == "id"][0]
print("join:", val_col, pmc_key, evt_key, name_col, d_evt, file=sys.stderr)
rows = cur.execute(f"""
SELECT k.display_name, c.{pmc_key}, SUM(c.{val_col})
FROM {cr} c JOIN {disp} d ON c.{evt_key} = d.{d_evt}
JOIN {sym} k ON d.kernel_id = k.id
GROUP BY k.display_name, c.{pmc_key}
""").fetchall()
names = dict(cur.execute(f"SELECT {ci_id}, {name_col} FROM {ci}"))
agg = {}
for kname, cid, val in rows:
    agg.setdefault(kname.split("(")[0][:60], {})[names.get(cid, cid)] = val
for k, d in sorted(agg.items(), key=lambda kv: -max(kv[1].values())):
    print(k)
    for n, v in sorted(d.items(), key=str):
        print(f"    {n}: {v:,.0f}")
