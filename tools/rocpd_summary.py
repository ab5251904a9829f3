#!/usr/bin/env python3
"""Summarise a rocprofv3 .db (rocpd) into a kernel-stats table.
Usage: python tools/rocpd_summary.py <results.db> [out.md]"""
import sqlite3
import sys


def summarise(path: str) -> str:
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = next(t for t in tables if t.startswith("rocpd_kernel_dispatch_"))
    sfx = sfx[len("rocpd_kernel_dispatch_"):]
    rows = cur.execute(f"""
      SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
             AVG(kd.end-kd.start)/1e6, MIN(kd.end-kd.start)/1e6,
             MAX(kd.grid_size_x*kd.grid_size_y*kd.grid_size_z)
      FROM rocpd_kernel_dispatch_{sfx} kd
      JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
      GROUP BY ks.display_name ORDER BY SUM(kd.end-kd.start) DESC
    """).fetchall()
    total = sum(r[2] for r in rows)
    out = ["| total ms | calls | avg ms | min ms | grid | % | kernel |",
           "|---|---|---|---|---|---|---|"]
    for r in rows[:20]:
        out.append(
            f"| {r[2]:.3f} | {r[1]} | {r[3]:.4f} | {r[4]:.4f} | {int(r[5])} "
            f"| {100*r[2]/total:.1f}% | `{r[0][:90]}` |"
        )
    out.append(f"\ntotal GPU kernel time: {total:.2f} ms")
    return "\n".join(out)


if __name__ == "__main__":
    text = summarise(sys.argv[1])
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")
    print(text)
