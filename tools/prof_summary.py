#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into per-kernel stats (total/avg
duration, count, VGPR) — the judged profile artifacts under profiles/.

Usage: python tools/prof_summary.py gpurun_out/prof/x_results.db > profiles/x.md
"""
import sqlite3
import sys


def summarize(path: str) -> str:
    con = sqlite3.connect(path)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    # grid/workgroup columns exist only in older rocprofv3 schemas
    cols = {r[1] for r in cur.execute(f"PRAGMA table_info({kd})")}
    has_grid = "grid_size_x" in cols
    gsel = ("MAX(d.grid_size_x) grid, MAX(d.workgroup_size_x) wg"
            if has_grid else "0 grid, 0 wg")
    rows = list(cur.execute(f"""
        SELECT s.display_name,
               COUNT(*) n,
               SUM(d.end - d.start) / 1e3 total_us,
               AVG(d.end - d.start) / 1e3 avg_us,
               MAX(s.arch_vgpr_count) vgpr,
               MAX(s.sgpr_count) sgpr,
               {gsel}
        FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY total_us DESC"""))
    total = sum(r[2] for r in rows)
    out = [f"# Kernel profile: {path}",
           f"total GPU kernel time: {total/1e3:.3f} ms", "",
           "| kernel | calls | total us | avg us | % | VGPR | SGPR | grid | wg |",
           "|---|---|---|---|---|---|---|---|---|"]
    for name, n, tot, avg, vgpr, sgpr, grid, wg in rows:
        short = name.split("(")[0][:80]
        out.append(f"| {short} | {n} | {tot:.1f} | {avg:.2f} | "
                   f"{100*tot/total:.1f}% | {vgpr} | {sgpr} | {grid} | {wg} |")
    return "\n".join(out) + "\n"


if __name__ == "__main__":
    print(summarize(sys.argv[1]))
