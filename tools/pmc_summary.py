#!/usr/bin/env python3
"""Summarise rocprofv3 --pmc results (rocpd SQLite .db output) into a
per-kernel DRAM-traffic table: dispatches, counter GB, Σduration,
effective rate.

rocprofv3 7.2 on gfx950 writes `<out>_results.db`; counters must be
collected ONE per run (FETCH_SIZE + WRITE_SIZE in one pass exceeds the
hardware counter capacity).  FETCH_SIZE/WRITE_SIZE are in KB
(documented expression: TCC_EA0 request counts × line size / 1024).

Usage: python tools/pmc_summary.py gpurun_out/pmc/sel_FETCH_SIZE_results.db
"""
import re
import sqlite3
import sys


def summarize(db):
    c = sqlite3.connect(db)
    names = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_pmc_event%'")]
    if not names:
        raise SystemExit(f"{db}: no rocpd_pmc_event table")
    sfx = names[0].replace('rocpd_pmc_event_', '')
    counter = c.execute(
        f"SELECT name FROM rocpd_info_pmc_{sfx}").fetchone()[0]
    q = f"""
    SELECT ks.display_name, count(*), sum(pe.value),
           sum(kd.end-kd.start)/1e3
    FROM rocpd_pmc_event_{sfx} pe
    JOIN rocpd_kernel_dispatch_{sfx} kd ON kd.event_id = pe.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON ks.id = kd.kernel_id
    GROUP BY ks.display_name ORDER BY 4 DESC
    """
    print(f"### {db} — counter {counter} (KB units)")
    print(f"| kernel | dispatches | {counter} GB | Σus | TB/s |")
    print("|---|---|---|---|---|")
    for name, n, val, us in c.execute(q):
        short = re.split(r'[<(]', name)[0].strip()[:48]
        gb = val / 1024 / 1024
        bw = gb / (us / 1e6) / 1024 if us else 0
        print(f"| {short} | {n} | {gb:.2f} | {us:.0f} | {bw:.2f} |")


if __name__ == "__main__":
    for db in sys.argv[1:]:
        summarize(db)
