#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a kernel-time markdown table.

Usage: python tools/kernel_stats.py <results.db> [top_n]
(rocprofv3 --kernel-trace --stats -d DIR -- <cmd> writes DIR/*/NNN_results.db)
"""

import sqlite3
import sys


def main():
    path = sys.argv[1]
    top_n = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = db.execute(
        f"""SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
                   AVG(d.end - d.start), MAX(s.arch_vgpr_count),
                   MAX(s.accum_vgpr_count), MAX(s.group_segment_size)
            FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"""
    ).fetchall()
    total_ns = sum(r[2] for r in rows)
    print("| kernel | calls | total_ms | avg_us | % | vgpr | agpr | lds |")
    print("|---|---|---|---|---|---|---|---|")
    for name, calls, tot, avg, vgpr, agpr, lds in rows[:top_n]:
        print(
            f"| {name[:70]} | {calls} | {tot / 1e6:.1f} | {avg / 1e3:.1f} "
            f"| {100.0 * tot / total_ns:.1f} | {vgpr} | {agpr} | {lds} |"
        )
    print(f"\nTotal GPU kernel time: {total_ns / 1e6:.1f} ms")


if __name__ == "__main__":
    main()
