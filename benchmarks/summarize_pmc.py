#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results .db per kernel.

Usage: python benchmarks/summarize_pmc.py <results.db>
Prints one line per kernel with ratio columns for whichever of the
known counters were collected.
"""
import collections
import sqlite3
import sys


def main(db_path: str):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [t for t in tables if t.startswith("rocpd_pmc_event_")][0][
        len("rocpd_pmc_event_"):]
    q = f"""
    SELECT s.display_name, p.name, SUM(e.value)
    FROM rocpd_pmc_event_{sfx} e
    JOIN rocpd_info_pmc_{sfx} p ON p.id = e.pmc_id
    JOIN rocpd_kernel_dispatch_{sfx} d ON d.id = e.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} s ON s.id = d.kernel_id
    GROUP BY s.display_name, p.name
    """
    agg = collections.defaultdict(dict)
    for name, ctr, val in cur.execute(q):
        agg[name.split("(")[0][:48]][ctr] = val

    mfma = "SQ_VALU_MFMA_BUSY_CYCLES"
    busy = "SQ_BUSY_CYCLES"
    wait = "SQ_WAIT_ANY"
    lds = "SQ_INSTS_LDS"
    conf = "SQ_LDS_BANK_CONFLICT"
    for k, v in sorted(agg.items(), key=lambda kv: -kv[1].get(busy, 0))[:12]:
        b = v.get(busy, 1) or 1
        cols = [f"{k:48s}", f"busy={b:.3g}"]
        if mfma in v:
            cols.append(f"mfma/busy={v[mfma] / b:6.3f}")
        if wait in v:
            cols.append(f"wait/busy={v[wait] / b:6.3f}")
        if conf in v and v.get(lds):
            cols.append(f"ldsconf/lds={v[conf] / v[lds]:6.3f}")
        print("  ".join(cols))


if __name__ == "__main__":
    main(sys.argv[1])
