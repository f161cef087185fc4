#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a kernel-time table (markdown)."""
import glob
import re
import sqlite3
import sys


def summarize(path, top=25):
    db = sqlite3.connect(path)
    cur = db.cursor()
    sfx = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")
        if r[0].startswith('rocpd_kernel_dispatch')][0].replace('rocpd_kernel_dispatch_', '')
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end - kd.start)
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC""").fetchall()
    total = sum(r[2] for r in rows)
    out = [f"total GPU kernel time: {total/1e9:.3f} s over "
           f"{sum(r[1] for r in rows)} dispatches\n",
           "| % | time (ms) | calls | kernel |", "|---|---|---|---|"]
    for name, n, t in rows[:top]:
        name = re.sub(r'\(.*', '', name).strip()[:80]
        out.append(f"| {t/total*100:.2f} | {t/1e6:.1f} | {n} | `{name}` |")
    return "\n".join(out)


if __name__ == "__main__":
    for arg in sys.argv[1:] or sorted(glob.glob("gpurun_out/prof*/*results.db")):
        print(f"\n## {arg}\n")
        print(summarize(arg))
