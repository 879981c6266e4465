#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a per-kernel table.

Usage: python tools/profile_summary.py gpurun_out/profX/bench_results.db [steps]
Writes markdown to stdout (redirect into profiles/).
"""
import sqlite3
import sys


def summarize(db_path: str, steps: int = 13):
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
    rows = cur.execute(q).fetchall()
    tot = sum(r[2] for r in rows)
    print(f"| kernel | calls | total ms | avg µs | % |")
    print(f"|---|---|---|---|---|")
    for name, n, ms, avg in rows[:25]:
        short = str(name).split("(")[0].replace("void ", "")[:70]
        print(f"| `{short}` | {n} | {ms:.2f} | {avg:.1f} | {100*ms/tot:.1f} |")
    print(f"\nTotal kernel time: {tot:.2f} ms over {steps} steps "
          f"= {tot/steps:.3f} ms/step GPU-busy")


if __name__ == "__main__":
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 13)
