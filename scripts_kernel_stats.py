"""Summarize a rocprofv3 rocpd .db: per-kernel call count + time.

Usage: python scripts_kernel_stats.py <results.db> [out.txt]
Handles the GUID-suffixed table names of rocprofv3 7.x rocpd databases.
"""

import sqlite3
import sys


def main():
    db = sys.argv[1]
    out_path = sys.argv[2] if len(sys.argv) > 2 else None
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]

    def find(prefix):
        for t in tables:
            if t.startswith(prefix):
                return t
        raise RuntimeError(f"table {prefix}* not found in {tables}")

    disp = find("rocpd_kernel_dispatch")
    sym = find("rocpd_info_kernel_symbol")
    q = f"""
    SELECT k.display_name AS name, COUNT(*) AS calls,
           SUM(d.end - d.start) AS total_ns,
           AVG(d.end - d.start) AS mean_ns
    FROM {disp} d
    JOIN {sym} k ON d.kernel_id = k.id
    GROUP BY name ORDER BY total_ns DESC LIMIT 45
    """
    lines = [f"{'calls':>8} {'total_ms':>12} {'mean_us':>10}  kernel"]
    total_gpu_ns = 0
    for name, calls, total_ns, mean_ns in cur.execute(q):
        total_gpu_ns += total_ns
        short = str(name).split("(")[0][:110]
        lines.append(f"{calls:>8} {total_ns / 1e6:>12.3f} {mean_ns / 1e3:>10.2f}  {short}")
    lines.insert(1, f"# total GPU kernel time: {total_gpu_ns / 1e6:.1f} ms")
    text = "\n".join(lines)
    print(text)
    if out_path:
        with open(out_path, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
