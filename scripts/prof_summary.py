#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db: top kernels by total time."""
import glob
import sqlite3
import sys

db_path = sorted(glob.glob(sys.argv[1]))[-1]
db = sqlite3.connect(db_path)
cur = db.cursor()
rows = list(cur.execute(
    "SELECT name, total_calls, total_duration, average, percentage "
    "FROM top_kernels LIMIT 25"))
for name, calls, tot, avg, pct in rows:
    short = name.split("(")[0][:90]
    print(f"{pct:5.1f}%  n={calls:6d}  avg={avg:9.1f}us  tot={tot/1e3:9.1f}ms  {short}")
