#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd DB: top kernels (and counters if present)."""
import glob
import sqlite3
import sys

pattern = sys.argv[1]
dbs = sorted(glob.glob(pattern))
if not dbs:
    print(f"no db matches {pattern}")
    sys.exit(1)
db = sqlite3.connect(dbs[-1])
tabs = [t[0] for t in db.execute(
    "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
if "top_kernels" in tabs:
    for r in db.execute("SELECT name, total_calls, average, percentage "
                        "FROM top_kernels LIMIT 18"):
        print(f"{r[3]:5.1f}%  x{r[1]:4d}  {r[2]:9.1f}us  {r[0][:96]}")
ctabs = [t for t in tabs if "counter" in t.lower()]
if ctabs:
    print("# counter tables:", ctabs[:4])
    try:
        cols = [d[1] for d in db.execute(f"PRAGMA table_info({ctabs[0]})")]
        print("# cols:", cols[:12])
    except Exception as e:
        print("ERR", e)
