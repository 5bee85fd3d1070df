#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results .db into per-kernel counter totals
(+ derived bandwidth when the counter is FETCH_SIZE/WRITE_SIZE in KB).

Usage: python tools/pmc_summary.py gpurun_out/pmc/x_results.db > profiles/x.md

NOTE (round 2): rocprofv3 --pmc hangs on the bench workloads on the
current ROCm 7.2 stack (two bounded-timeout attempts — see
profiles/mf32_r02_final.md); the round-1 PMC artifacts in profiles/
were produced with this tool when collection still worked.
"""
import sqlite3
import sys


def summarize(path: str) -> str:
    con = sqlite3.connect(path)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    ev = next(t for t in tabs if t.startswith("rocpd_pmc_event"))
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    info = next(t for t in tabs if t.startswith("rocpd_info_pmc"))
    counters = {i: n for i, n in cur.execute(
        f"SELECT id, name FROM {info}")}
    out = [f"# PMC counter profile: {path}",
           f"counters: {sorted(set(counters.values()))}", ""]
    rows = list(cur.execute(f"""
        SELECT s.display_name, e.pmc_id, COUNT(*), SUM(e.value),
               SUM(d.end - d.start) / 1e3
        FROM {ev} e
        JOIN {kd} d ON e.event_id = d.event_id
        JOIN {ks} s ON d.kernel_id = s.id
        GROUP BY s.display_name, e.pmc_id
        ORDER BY SUM(e.value) DESC LIMIT 20"""))
    out.append("| kernel | counter | dispatches | total | GB | eff GB/s |")
    out.append("|---|---|---|---|---|---|")
    for name, pmc, n, val, us in rows:
        short = name.split("(")[0][:60]
        gb = val / 1024 / 1024  # counters are KB
        bw = gb / (us / 1e6) if us else 0.0
        out.append(f"| {short} | {counters.get(pmc, pmc)} | {n} | "
                   f"{val:.0f} | {gb:.2f} | {bw:.0f} |")
    return "\n".join(out) + "\n"


if __name__ == "__main__":
    sys.stdout.write(summarize(sys.argv[1]))
