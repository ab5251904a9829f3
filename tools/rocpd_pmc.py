#!/usr/bin/env python3
"""Summarise PMC counters from a rocprofv3 rocpd db, per kernel."""
import sqlite3, sys
db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = next(t for t in tables if t.startswith("rocpd_kernel_dispatch_"))[len("rocpd_kernel_dispatch_"):]
try:
    rows = cur.execute(f"""
      SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
      FROM rocpd_pmc_event_{sfx} pe
      JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
      JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id = kd.event_id
      JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
      GROUP BY ks.display_name, pi.name ORDER BY ks.display_name""").fetchall()
except Exception as e:
    print("query failed:", e)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info(rocpd_pmc_event_{sfx})")]
    print("pmc_event cols:", cols)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info(rocpd_info_pmc_{sfx})")]
    print("info_pmc cols:", cols)
    sys.exit(1)
for name, ctr, val, n in rows:
    if "ffa" in name or "at::" not in name:
        print(f"{ctr:32s} {val:>18,.0f}  ({n} disp)  {name[:60]}")
