#!/usr/bin/env python3
"""Aggregate rocprofv3 PMC counters per kernel from a rocpd sqlite db.

Usage: python scripts/rocpd_pmc.py <db> [out.md]
Prints per-kernel counter sums plus derived MFMA utilization
(SQ_VALU_MFMA_BUSY_CYCLES / (SQ_BUSY_CYCLES * 4 SIMD)) and LDS-conflict
rate where the counters are present.
"""

import sqlite3
import sys
from collections import defaultdict


def main(db_path, out=None):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    u = None
    for (name,) in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        if name.startswith("rocpd_pmc_event_"):
            u = name[len("rocpd_pmc_event_"):]
    assert u
    # pmc_id -> counter name
    pmc_names = dict(cur.execute(
        f"SELECT id, name FROM rocpd_info_pmc_{u}").fetchall())
    # event_id -> kernel name via kernel_dispatch
    q = f"""
    SELECT ks.display_name, pi.name, SUM(pe.value)
    FROM rocpd_pmc_event_{u} pe
    JOIN rocpd_kernel_dispatch_{u} kd ON pe.event_id = kd.event_id
    JOIN rocpd_info_kernel_symbol_{u} ks ON kd.kernel_id = ks.id
    JOIN rocpd_info_pmc_{u} pi ON pe.pmc_id = pi.id
    WHERE ks.display_name LIKE 'drla%'
    GROUP BY ks.display_name, pi.name
    """
    agg = defaultdict(dict)
    for kname, cname, val in cur.execute(q):
        agg[kname][cname] = val
    lines = [
        "PMC run: scripts/conv_microbench.py under rocprofv3 --pmc "
        "(counters-only run). MFMA util = SQ_VALU_MFMA_BUSY_CYCLES / "
        "(SQ_BUSY_CYCLES * 4 SIMD), i.e. matrix-pipe busy fraction while "
        "the kernel occupies the SEs.",
        "",
        "| kernel | MFMA util % | LDS bank-conflict | MFMA insts | "
        "VALU insts |", "|---|---|---|---|---|"]
    for kname in sorted(agg):
        c = agg[kname]
        busy = c.get("SQ_BUSY_CYCLES", 0)
        mfma_busy = c.get("SQ_VALU_MFMA_BUSY_CYCLES", 0)
        util = 100.0 * mfma_busy / (busy * 4) if busy else 0.0
        conflicts = c.get("SQ_LDS_BANK_CONFLICT", 0)
        idx = c.get("SQ_LDS_IDX_ACTIVE")
        conf = (f"{100.0*conflicts/idx:6.2f}" if idx
                else f"{conflicts:.3g} cyc")
        lines.append(
            f"| `{kname[:48]}` | {util:6.1f} | {conf} | "
            f"{c.get('SQ_INSTS_MFMA', 0):.3g} | "
            f"{c.get('SQ_INSTS_VALU', 0):.3g} |")
    text = "\n".join(lines)
    if out:
        with open(out, "w") as f:
            f.write(text + "\n")
    print(text)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
