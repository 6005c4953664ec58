#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a per-step kernel
table (the profiles/*.md tables are generated with this).

Usage:
    python scripts/analyze_rocpd.py <results.db> [--steps N] [--top K]
                                    [--markdown]

`--steps` divides totals into per-step numbers (pass warmup+timed steps
of the profiled run).  rocprofv3 writes the db under
`<out-dir>/runc/<pid>_results.db`.
"""

from __future__ import annotations

import argparse
import sqlite3


def load_kernel_stats(db_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    rows = cur.execute(
        "SELECT k.display_name, COUNT(*), SUM(d.end - d.start) / 1e6, "
        "       AVG(d.end - d.start) / 1e3 "
        "FROM rocpd_kernel_dispatch d "
        "JOIN rocpd_info_kernel_symbol k ON d.kernel_id = k.id "
        "GROUP BY k.display_name ORDER BY 3 DESC").fetchall()
    total_ms, total_n = cur.execute(
        "SELECT SUM(end - start) / 1e6, COUNT(*) "
        "FROM rocpd_kernel_dispatch").fetchone()
    db.close()
    return rows, total_ms, total_n


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--steps", type=int, default=1,
                    help="steps in the profiled run (warmup + timed)")
    ap.add_argument("--top", type=int, default=25)
    ap.add_argument("--markdown", action="store_true")
    args = ap.parse_args()

    rows, total_ms, total_n = load_kernel_stats(args.db)
    s = args.steps
    print(f"total GPU busy {total_ms:.2f} ms / {total_n} dispatches"
          + (f" / {s} steps = {total_ms / s * 1000:.0f} us/step"
             if s > 1 else ""))
    if args.markdown:
        print("\n| kernel | calls/step | us/step | avg us |")
        print("|---|---|---|---|")
        for name, n, ms, avg in rows[:args.top]:
            nm = name.replace("|", "\\|")[:64]
            print(f"| `{nm}` | {n / s:.1f} | {ms / s * 1000:.1f} "
                  f"| {avg:.1f} |")
    else:
        for name, n, ms, avg in rows[:args.top]:
            print(f"{n / s:6.1f}/step  {ms / s * 1000:9.1f} us/step  "
                  f"{avg:8.1f} us avg  {name[:60]}")


if __name__ == "__main__":
    main()
