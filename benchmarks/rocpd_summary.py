#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite database into a kernel-stats table
(name, calls, total µs, mean µs, VGPR/SGPR). Usage:
    python benchmarks/rocpd_summary.py <results.db> [out.md]
"""

import sqlite3
import sys


def summarize(db_path: str) -> str:
    db = sqlite3.connect(db_path)
    tables = [r[0] for r in db.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = db.execute(
        f"""
        SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1000.0,
               AVG(d.end-d.start)/1000.0, MAX(s.arch_vgpr_count),
               MAX(s.accum_vgpr_count), MAX(s.sgpr_count), MAX(s.group_segment_size)
        FROM {disp} d JOIN {sym} s ON s.id = d.kernel_id
        GROUP BY s.display_name ORDER BY 3 DESC LIMIT 30
        """
    ).fetchall()
    total = sum(r[2] for r in rows)
    out = ["| kernel | calls | total µs | mean µs | % | vgpr | agpr | sgpr | lds |",
           "|---|---|---|---|---|---|---|---|---|"]
    for name, calls, tot, mean, vgpr, agpr, sgpr, lds in rows:
        short = name.split("(")[0][:80]
        out.append(
            f"| {short} | {calls} | {tot:.1f} | {mean:.2f} | {100*tot/total:.1f}% | {vgpr} | {agpr} | {sgpr} | {lds} |"
        )
    # memory copies if present
    try:
        mc = next(t for t in tables if t.startswith("rocpd_memory_copy"))
        r = db.execute(f"SELECT COUNT(*), SUM(end-start)/1000.0, SUM(size)/1.0e9 FROM {mc}").fetchone()
        if r and r[0]:
            out.append("")
            out.append(f"memory copies: {r[0]} totalling {r[1]:.1f} µs, {r[2]:.2f} GB")
    except (StopIteration, sqlite3.OperationalError):
        pass
    return "\n".join(out)


if __name__ == "__main__":
    text = summarize(sys.argv[1])
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")
    print(text)
