#!/usr/bin/env python3
"""Summarize a rocprofv3 results database: top kernels by total time.

Usage: python tools/prof_summary.py <results.db> [out.csv]
"""
import csv
import sqlite3
import sys


def main():
    db = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else None
    con = sqlite3.connect(db)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")]
    if not tabs:
        print("no kernel dispatch table found")
        return
    sfx = tabs[0][len("rocpd_kernel_dispatch_"):]
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6, AVG(kd.end-kd.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 30
    """
    rows = cur.execute(q).fetchall()
    tot = sum(r[2] for r in rows)
    w = None
    if out:
        f = open(out, "w", newline="")
        w = csv.writer(f)
        w.writerow(["kernel", "calls", "total_ms", "avg_us", "pct"])
    for name, calls, ms, us in rows:
        line = f"{100*ms/tot:5.1f}%  {ms:9.2f}ms  {calls:6d}x  {us:9.1f}us  {name[:90]}"
        print(line)
        if w:
            w.writerow([name[:140], calls, f"{ms:.3f}", f"{us:.2f}", f"{100*ms/tot:.1f}"])
    print(f"total kernel time: {tot:.1f} ms")


if __name__ == "__main__":
    main()
