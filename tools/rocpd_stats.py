"""Aggregate a rocprofv3 rocpd sqlite DB into a per-kernel stats CSV.

rocprofv3 (ROCm 7.2) writes tables with or without a per-session GUID suffix;
this resolves them dynamically.

    python tools/rocpd_stats.py <results.db> <out.csv> [top_n]
"""

import csv
import sqlite3
import sys


def main():
    db, out = sys.argv[1], sys.argv[2]
    top = int(sys.argv[3]) if len(sys.argv) > 3 else 60
    con = sqlite3.connect(db)
    tabs = [r[0] for r in
            con.execute("select name from sqlite_master where type='table'")]

    def tab(prefix):
        m = [t for t in tabs if t == prefix or t.startswith(prefix + "_0")]
        if not m:
            m = [t for t in tabs if t.startswith(prefix)]
        return m[0]

    kd, ks = tab("rocpd_kernel_dispatch"), tab("rocpd_info_kernel_symbol")
    name_col = "display_name"
    cols = [c[1] for c in con.execute(f"pragma table_info({ks})")]
    if name_col not in cols:
        name_col = "kernel_name"
    rows = con.execute(f"""
        select s.{name_col}, count(*) n, sum(d.end - d.start)/1e6 total_ms,
               avg(d.end - d.start)/1e3 avg_us
        from {kd} d join {ks} s on d.kernel_id = s.id
        group by s.{name_col} order by total_ms desc limit {top}""").fetchall()
    grand = con.execute(
        f"select sum(end - start)/1e6 from {kd}").fetchone()[0] or 0.0
    with open(out, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["kernel", "calls", "total_ms", "avg_us", "pct_gpu"])
        for name, n, tot, avg in rows:
            w.writerow([name[:140], n, round(tot, 2), round(avg, 2),
                        round(100.0 * tot / grand, 2) if grand else 0])
    for name, n, tot, avg in rows[:16]:
        print(f"{tot:9.1f} ms {n:6d}x {avg:9.1f} us  {name[:80]}")
    print(f"total GPU time: {grand:.1f} ms")


if __name__ == "__main__":
    main()
