#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite results DB into a markdown kernel table.

Usage: python tools/prof_summary.py <results.db> [out.md]
"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    out = open(sys.argv[2], "w") if len(sys.argv) > 2 else sys.stdout
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    print(f"# rocprofv3 kernel summary — {path}\n", file=out)
    print("| kernel | dispatches | total ms | avg us | max us | avg grid |",
          file=out)
    print("|---|---|---|---|---|---|", file=out)
    q = f"""
    SELECT ks.kernel_name, COUNT(*), SUM(k.end-k.start)/1e6,
           AVG(k.end-k.start)/1e3, MAX(k.end-k.start)/1e3, AVG(k.grid_size_x)
    FROM {disp} k JOIN {sym} ks ON k.kernel_id = ks.id
    GROUP BY ks.kernel_name ORDER BY 3 DESC LIMIT 25
    """
    for name, cnt, tot, avg, mx, grid in db.execute(q):
        name = name.replace(".kd", "").replace("_ZN5auron", "auron::")[:60]
        print(f"| `{name}` | {cnt} | {tot:.2f} | {avg:.1f} | {mx:.1f} | "
              f"{int(grid)} |", file=out)
    span = db.execute(
        f"SELECT (MAX(end)-MIN(start))/1e6 FROM {disp}").fetchone()[0]
    print(f"\nGPU kernel-activity span: {span:.1f} ms", file=out)


if __name__ == "__main__":
    main()
