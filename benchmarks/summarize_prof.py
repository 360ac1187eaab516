#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into a per-kernel table (markdown).

Usage: python benchmarks/summarize_prof.py <results.db> [out.md]
"""
import sqlite3
import sys
from pathlib import Path


def summarize(db_path: str):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
           AVG(d.end-d.start)/1e3, MIN(d.end-d.start)/1e3,
           AVG(d.grid_size_x*d.grid_size_y*d.grid_size_z)
    FROM {kd} d JOIN {ks} s ON s.id = d.kernel_id
    GROUP BY s.display_name ORDER BY 3 DESC
    """
    rows = list(cur.execute(q))
    total = sum(r[2] for r in rows)
    lines = [
        "| kernel | calls | total ms | avg us | min us | avg grid | % |",
        "|---|---|---|---|---|---|---|",
    ]
    for name, calls, ms, avg, mn, grid in rows:
        short = name.split("(")[0][:70]
        lines.append(
            f"| `{short}` | {calls} | {ms:.3f} | {avg:.2f} | {mn:.2f} "
            f"| {grid/256:.0f} wg | {100*ms/total:.1f} |")
    lines.append(f"\nTotal GPU kernel time: {total:.2f} ms")
    return "\n".join(lines)


if __name__ == "__main__":
    out = summarize(sys.argv[1])
    if len(sys.argv) > 2:
        Path(sys.argv[2]).write_text(out + "\n")
    print(out)
