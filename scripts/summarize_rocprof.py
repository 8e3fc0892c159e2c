#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB into a compact per-kernel table."""
import glob
import sqlite3
import sys


def summarize(db_glob, out_path, top=40):
    paths = glob.glob(db_glob)
    assert paths, f"no db matches {db_glob}"
    db = sqlite3.connect(paths[0])
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'"
    )][0]
    sfx = t.replace("rocpd_kernel_dispatch_", "")
    rows = cur.execute(
        f"""SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
            AVG(k.end-k.start)/1e3
            FROM rocpd_kernel_dispatch_{sfx} k
            JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id = s.id
            GROUP BY s.display_name ORDER BY 3 DESC LIMIT {top}"""
    ).fetchall()
    total = cur.execute(
        f"SELECT COUNT(*), SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{sfx}"
    ).fetchone()
    swq = cur.execute(
        f"""SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
            AVG(k.end-k.start)/1e3
            FROM rocpd_kernel_dispatch_{sfx} k
            JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id = s.id
            WHERE s.display_name LIKE 'swq_%'
            GROUP BY s.display_name ORDER BY 3 DESC"""
    ).fetchall()
    with open(out_path, "w") as f:
        f.write("| total_ms | count | avg_us | kernel |\n|---|---|---|---|\n")
        for name, cnt, ms, avg in rows:
            f.write(f"| {ms:.2f} | {cnt} | {avg:.1f} | {name[:100]} |\n")
        f.write(f"\nall kernels: {total[0]} dispatches, {total[1]:.1f} ms\n")
        f.write("\n## shockwave_amd HIP kernels (swq_*)\n\n")
        f.write("| total_ms | count | avg_us | kernel |\n|---|---|---|---|\n")
        for name, cnt, ms, avg in swq:
            f.write(f"| {ms:.2f} | {cnt} | {avg:.1f} | {name} |\n")
    print(f"wrote {out_path}")


if __name__ == "__main__":
    summarize(sys.argv[1], sys.argv[2])
