#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite database into per-kernel stats.

Usage: python scripts/rocpd_stats.py gpurun_out/prof/bench_results.db [out.md]
"""

import sqlite3
import sys


def summarize(db_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    uuid = None
    for (name,) in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        if name.startswith("rocpd_kernel_dispatch_"):
            uuid = name[len("rocpd_kernel_dispatch_"):]
    assert uuid, "no kernel dispatch table"
    q = f"""
    SELECT ks.display_name AS kernel,
           COUNT(*) AS calls,
           SUM(kd.end - kd.start) / 1e3 AS total_us,
           AVG(kd.end - kd.start) / 1e3 AS avg_us
    FROM rocpd_kernel_dispatch_{uuid} kd
    JOIN rocpd_info_kernel_symbol_{uuid} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name
    ORDER BY total_us DESC
    """
    rows = cur.execute(q).fetchall()
    total = sum(r[2] for r in rows)
    out = []
    out.append(f"total GPU kernel time: {total/1e3:.3f} ms over "
               f"{sum(r[1] for r in rows)} dispatches\n")
    out.append("| % | total us | calls | avg us | kernel |")
    out.append("|---|---|---|---|---|")
    for kernel, calls, tot_us, avg_us in rows[:40]:
        kshort = kernel if len(kernel) < 110 else kernel[:107] + "..."
        out.append(f"| {100*tot_us/total:5.1f} | {tot_us:10.1f} | {calls:5d} "
                   f"| {avg_us:8.2f} | `{kshort}` |")
    return "\n".join(out)


if __name__ == "__main__":
    text = summarize(sys.argv[1])
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")
    print(text)
