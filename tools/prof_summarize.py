#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB to a small text file, for gpurun copy-back."""
import sqlite3, sys
db, out = sys.argv[1], sys.argv[2]
con = sqlite3.connect(db)
tabs = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
with open(out, "w") as f:
    if "top_kernels" in tabs:
        rows = list(con.execute("SELECT name,total_calls,total_duration,average,percentage FROM top_kernels ORDER BY total_duration DESC LIMIT 40"))
        f.write("name|calls|total_us|avg_us|pct\n")
        for n,c,d,a,p in rows:
            f.write(f"{n.split('(')[0][:100]}|{c}|{d:.1f}|{a:.2f}|{p:.3f}\n")
    try:
        cols = [r[1] for r in con.execute("PRAGMA table_info(counters_collection)")]
        f.write(f"\n== pmc cols: {cols}\n")
        rows = list(con.execute("""
            SELECT kernel_name, counter_name, SUM(value), COUNT(*)
            FROM counters_collection GROUP BY 1,2 ORDER BY 3 DESC LIMIT 300"""))
        f.write("kernel|counter|sum|n\n")
        for k,cn,v,n in rows:
            f.write(f"{str(k).split('(')[0][:80]}|{cn}|{v:.0f}|{n}\n")
    except Exception as e:
        f.write(f"\n(no pmc aggregation: {e})\n")
        f.write("tables: " + ",".join(tabs) + "\n")
print("wrote", out)
