#!/usr/bin/env python3
"""Summarize rocprofv3 output into a per-kernel markdown table.

Accepts either a rocpd SQLite database (default output format) or a
`*_kernel_stats.csv` from `rocprofv3 --stats --output-format csv`.

Usage: python tools/profile_summary.py <results.db | ..._kernel_stats.csv> [steps]
Writes markdown to stdout (redirect into profiles/).
"""
import sqlite3
import sys


def summarize_csv(csv_path: str, steps: int = 13):
    import csv as _csv
    recs = list(_csv.DictReader(open(csv_path)))
    rows = sorted(((r["Name"], int(r["Calls"]),
                    float(r["TotalDurationNs"]) / 1e6,
                    float(r["AverageNs"]) / 1e3) for r in recs),
                  key=lambda r: -r[2])
    _emit(rows, steps)


def _emit(rows, steps):
    tot = sum(r[2] for r in rows)
    print("| kernel | calls | total ms | avg µs | % |")
    print("|---|---|---|---|---|")
    for name, n, ms, avg in rows[:25]:
        short = str(name).split("(")[0].replace("void ", "")[:70]
        print(f"| `{short}` | {n} | {ms:.2f} | {avg:.1f} | {100*ms/tot:.1f} |")
    print(f"\nTotal kernel time: {tot:.2f} ms over {steps} steps "
          f"= {tot/steps:.3f} ms/step GPU-busy")


def summarize(db_path: str, steps: int = 13):
    if db_path.endswith(".csv"):
        summarize_csv(db_path, steps)
        return
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sfx = kd.replace("rocpd_kernel_dispatch", "")
    q = f"""SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms,
                   AVG(k.end-k.start)/1e3 avg_us
            FROM rocpd_kernel_dispatch{sfx} k
            JOIN rocpd_info_kernel_symbol{sfx} ks ON k.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY ms DESC"""
    rows = [(name, n, ms, avg) for name, n, ms, avg in cur.execute(q).fetchall()]
    _emit(rows, steps)


if __name__ == "__main__":
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 13)
