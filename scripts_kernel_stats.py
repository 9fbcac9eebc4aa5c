"""Summarize a rocprofv3 rocpd .db: per-kernel call count + time.

Usage: python scripts_kernel_stats.py <results.db> [out.txt]
"""

import sqlite3
import sys


def main():
    db = sys.argv[1]
    out_path = sys.argv[2] if len(sys.argv) > 2 else None
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'"
    )]
    # rocpd schema: kernel dispatch rows reference kernel name via string table
    q = None
    if "rocpd_kernel_dispatch" in tables:
        cols = [r[1] for r in cur.execute("PRAGMA table_info(rocpd_kernel_dispatch)")]
        # find plausible column names
        name_join = None
        if "kernel_id" in cols and "rocpd_info_kernel_symbol" in tables:
            kcols = [r[1] for r in cur.execute("PRAGMA table_info(rocpd_info_kernel_symbol)")]
            sym = "display_name" if "display_name" in kcols else (
                "kernel_name" if "kernel_name" in kcols else kcols[1]
            )
            q = f"""
            SELECT k.{sym} AS name, COUNT(*) AS calls,
                   SUM(d.end - d.start) AS total_ns,
                   AVG(d.end - d.start) AS mean_ns
            FROM rocpd_kernel_dispatch d
            JOIN rocpd_info_kernel_symbol k ON d.kernel_id = k.id
            GROUP BY name ORDER BY total_ns DESC LIMIT 40
            """
    if q is None:
        print("tables:", tables)
        for t in tables:
            cols = [r[1] for r in cur.execute(f"PRAGMA table_info({t})")]
            print(t, cols)
        return
    lines = [f"{'calls':>8} {'total_ms':>12} {'mean_us':>10}  kernel"]
    for name, calls, total_ns, mean_ns in cur.execute(q):
        short = str(name).split("(")[0][:100]
        lines.append(f"{calls:>8} {total_ns / 1e6:>12.3f} {mean_ns / 1e3:>10.2f}  {short}")
    text = "\n".join(lines)
    print(text)
    if out_path:
        with open(out_path, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
