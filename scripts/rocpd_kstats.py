#!/usr/bin/env python3
"""Aggregate per-kernel time from a rocprofv3 rocpd SQLite database
(this ROCm's rocprofv3 emits only the DB). Usage: rocpd_kstats.py DB"""
import sqlite3
import sys

conn = sqlite3.connect(sys.argv[1])
tables = [r[0] for r in conn.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]
disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
rows = conn.execute(f"""
  SELECT s.kernel_name, COUNT(*), SUM(d.end - d.start),
         AVG(d.end - d.start)
  FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
  GROUP BY s.kernel_name ORDER BY SUM(d.end - d.start) DESC
""").fetchall()
total = sum(r[2] for r in rows)
print(f"{'kernel':58s} {'calls':>7s} {'total_ms':>9s} {'avg_us':>8s} {'%':>5s}")
for name, calls, tot, avg in rows[:28]:
    nm = name
    for pre in ("_ZN12_GLOBAL__N_1", "void "):
        if nm.startswith(pre):
            nm = nm[len(pre):]
    nm = nm.split("(")[0][:58]
    print(f"{nm:58s} {calls:7d} {tot/1e6:9.2f} {avg/1e3:8.1f} "
          f"{100*tot/total:5.1f}")
print(f"TOTAL kernel time: {total/1e6:.2f} ms")
