#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a per-kernel table
(total/avg time, launches, grid, VGPR). Usage:

    python gpu_scripts/prof_summary.py gpurun_out/prof1/prof1_results.db [out.md]
"""

import re
import sqlite3
import sys


def summarize(db_path, out_path=None):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
               AVG(d.end - d.start), AVG(d.grid_size_x * d.grid_size_y *
               d.grid_size_z), MAX(s.arch_vgpr_count), MAX(s.accum_vgpr_count),
               MAX(d.group_segment_size)
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC
    """).fetchall()
    total_ns = sum(r[2] for r in rows)
    lines = ["| kernel | calls | total ms | avg µs | %GPU | avg grid | VGPR+ACC | LDS |",
             "|---|---|---|---|---|---|---|---|"]
    for name, calls, tot, avg, grid, vgpr, acc, lds in rows[:40]:
        short = re.sub(r"<[^<>]*>", "", name)[:60]
        lines.append(
            f"| {short} | {calls} | {tot / 1e6:.2f} | {avg / 1e3:.1f} | "
            f"{100 * tot / total_ns:.1f} | {grid:.0f} | {vgpr}+{acc} | {lds} |")
    lines.append(f"\nTotal GPU kernel time: {total_ns / 1e6:.2f} ms "
                 f"across {sum(r[1] for r in rows)} dispatches")
    text = "\n".join(lines)
    if out_path:
        with open(out_path, "w") as f:
            f.write(text + "\n")
    print(text)


if __name__ == "__main__":
    summarize(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
