#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB into a small per-kernel CSV/markdown
(run on the GPU box so only the summary travels back)."""

from __future__ import annotations

import re
import sqlite3
import sys


def main(db_path: str, out_path: str, top: int = 50):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3
        FROM {kd} k JOIN {ks} s ON k.kernel_id = s.id
        GROUP BY s.display_name ORDER BY 3 DESC LIMIT {top}
    """).fetchall()
    total = cur.execute(f"SELECT SUM(end-start)/1e6 FROM {kd}").fetchone()[0]
    n = cur.execute(f"SELECT COUNT(*) FROM {kd}").fetchone()[0]
    span = cur.execute(
        f"SELECT (MAX(end)-MIN(start))/1e6 FROM {kd}").fetchone()[0]
    with open(out_path, "w") as f:
        f.write(f"# total_kernel_ms={total:.0f} dispatches={n} "
                f"wall_span_ms={span:.0f}\n")
        f.write("ms,pct,count,avg_us,kernel\n")
        for name, cnt, ms, avg_us in rows:
            short = re.sub(r"\(.*", "", name).replace("void ", "")[:110]
            f.write(f"{ms:.1f},{100*ms/total:.1f},{cnt},{avg_us:.1f},{short}\n")
    print(f"wrote {out_path}: total {total:.0f} ms, {n} dispatches, "
          f"span {span:.0f} ms")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else "prof_summary.csv")
