#!/usr/bin/env python3
"""Summarise a rocprofv3 rocpd SQLite database into per-kernel stats CSV
(the kernel-stats workflow changed to .db output in ROCm 7.2's rocprofv3).

Usage: python tools/prof_stats.py <results.db> [out.csv] [--top N]
Prints a ranked table and optionally writes the full CSV.
"""

import sqlite3
import sys


def main():
    db = sys.argv[1]
    out_csv = None
    top = 25
    args = sys.argv[2:]
    while args:
        a = args.pop(0)
        if a == "--top":
            top = int(args.pop(0))
        else:
            out_csv = a
    c = sqlite3.connect(db)
    tables = [r[0] for r in c.execute(
        "select name from sqlite_master where type='table'").fetchall()]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = c.execute(f"""
        select s.display_name, count(*), sum(d.end - d.start),
               avg(d.end - d.start), min(d.end - d.start), max(d.end - d.start)
        from {disp} d join {sym} s on d.kernel_id = s.id
        group by s.display_name order by sum(d.end - d.start) desc
    """).fetchall()
    total = sum(r[2] for r in rows)
    print(f"{'%':>6} {'calls':>7} {'total_us':>11} {'avg_us':>9}  name")
    for name, calls, tot, avg, mn, mx in rows[:top]:
        print(f"{100.0 * tot / total:6.2f} {calls:7d} {tot / 1e3:11.1f} "
              f"{avg / 1e3:9.2f}  {name[:110]}")
    print(f"total kernel time: {total / 1e6:.3f} ms over {sum(r[1] for r in rows)} dispatches")
    if out_csv:
        with open(out_csv, "w") as f:
            f.write('"Name","Calls","TotalDurationNs","AverageNs","Percentage","MinNs","MaxNs"\n')
            for name, calls, tot, avg, mn, mx in rows:
                f.write(f'"{name}",{calls},{tot},{avg:.1f},{100.0 * tot / total:.4f},{mn},{mx}\n')
        print(f"wrote {out_csv}")


if __name__ == "__main__":
    main()
