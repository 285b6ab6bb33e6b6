#!/usr/bin/env python3
"""Distill rocprofv3 rocpd .db output (gpurun_out/) into the committed
text summaries in profiles/.  Usage: summarize.py <db> [label]"""
import sqlite3, sys

def open_db(path):
    con = sqlite3.connect(path)
    cur = con.cursor()
    for (n,) in cur.execute("SELECT name FROM sqlite_master WHERE type='table'"):
        if n.startswith('rocpd_kernel_dispatch_'):
            return con, n[len('rocpd_kernel_dispatch_'):]
    raise SystemExit("no kernel dispatch table")

con, sfx = open_db(sys.argv[1])
cur = con.cursor()
print(f"{'kernel':60s} {'n':>4s} {'total_ms':>9s} {'avg_ms':>8s} {'vgpr':>5s} grid")
for r in cur.execute(f"""
SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6, AVG(kd.end-kd.start)/1e6,
       MAX(ks.arch_vgpr_count), MAX(kd.grid_size_x), MAX(kd.grid_size_y)
FROM rocpd_kernel_dispatch_{sfx} kd
JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
GROUP BY 1 ORDER BY 3 DESC"""):
    print(f"{r[0][:60]:60s} {r[1]:4d} {r[2]:9.3f} {r[3]:8.3f} {r[4]:5d} {r[5]}x{r[6]}")
try:
    rows = list(cur.execute(f"""
SELECT ks.display_name, pi.name, SUM(pe.value)
FROM rocpd_pmc_event_{sfx} pe
JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id=kd.event_id
JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id=pi.id GROUP BY 1,2"""))
    if rows:
        print("\nPMC (summed over dispatches and SQ/TCC instances):")
        for name, pmc, val in rows:
            print(f"  {name[:52]:52s} {pmc:22s} {val:.4e}")
except Exception:
    pass
