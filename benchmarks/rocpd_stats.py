#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a per-kernel stats table
(total ms, calls, avg us, % of GPU time).  Usage: rocpd_stats.py <db> [topN]"""
import sqlite3
import sys


def summarize(path, top=25):
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = db.execute(f"""
        SELECT ks.display_name AS name, COUNT(*) AS calls,
               SUM(k.end-k.start)/1e6 AS ms, AVG(k.end-k.start)/1e3 AS us
        FROM {kd} k JOIN {ks} ks ON k.kernel_id = ks.id
        GROUP BY name ORDER BY ms DESC""").fetchall()
    total = sum(r[2] for r in rows)
    out = [f"total GPU kernel time: {total:.2f} ms over "
           f"{sum(r[1] for r in rows)} dispatches", ""]
    out.append(f"{'ms':>10} {'%':>6} {'calls':>7} {'avg us':>9}  name")
    for name, calls, ms, us in rows[:top]:
        out.append(f"{ms:10.3f} {100*ms/total:6.1f} {calls:7d} {us:9.1f}  {name[:100]}")
    return "\n".join(out)


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 25
    print(summarize(sys.argv[1], top))
