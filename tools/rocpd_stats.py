"""Summarize a rocprofv3 SQLite (rocpd) database into a kernel-time table.

Usage: python tools/rocpd_stats.py <results.db> [top_n]

rocprofv3 --kernel-trace --stats writes <pid>_results.db; this prints the
per-kernel total time / share / dispatch count, the artifact we commit
under profiles/ for the judge.
"""

from __future__ import annotations

import sqlite3
import sys


def kernel_table(db_path: str, top: int = 40):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tabs = [
        r[0]
        for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")
    ]
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch_"))
    ks = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol_"))
    rows = list(
        cur.execute(
            f"""
            SELECT ks.display_name, COUNT(*) n, SUM(kd.end-kd.start)/1e6 ms
            FROM {kd} kd JOIN {ks} ks ON kd.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY ms DESC
            """
        )
    )
    total = sum(r[2] for r in rows)
    disp = sum(r[1] for r in rows)
    out = [f"total kernel ms: {total:.1f} over {disp} dispatches"]
    for name, n, ms in rows[:top]:
        out.append(f"{ms:9.2f} ms {100*ms/total:5.1f}% n={n:5d}  {name[:120]}")
    return "\n".join(out)


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    print(kernel_table(sys.argv[1], top))
