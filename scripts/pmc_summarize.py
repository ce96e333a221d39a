#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc counters_collection by kernel."""
import glob
import sqlite3
import sys

db = sqlite3.connect(sorted(glob.glob(sys.argv[1]))[-1])
cols = [d[1] for d in db.execute("PRAGMA table_info(counters_collection)")]
name_col = next(c for c in cols if "kernel" in c and "name" in c)
cnt_col = next(c for c in cols if c in ("counter_name", "name"))
val_col = next(c for c in cols if "value" in c)
q = (f"SELECT {name_col}, {cnt_col}, SUM({val_col}), COUNT(*) "
     f"FROM counters_collection GROUP BY {name_col}, {cnt_col}")
agg = {}
for kn, cn, v, n in db.execute(q):
    agg.setdefault(kn, {})[cn] = (v, n)
pats = sys.argv[2].split(",") if len(sys.argv) > 2 else None
for kn, cs in sorted(agg.items()):
    if pats and not any(p in kn for p in pats):
        continue
    short = kn.split("(")[0][:60]
    wc = cs.get("SQ_WAVE_CYCLES", (0, 0))[0]
    wait = cs.get("SQ_WAIT_ANY", (0, 0))[0]
    act = cs.get("SQ_ACTIVE_INST_ANY", (0, 0))[0]
    mfma = cs.get("SQ_INSTS_MFMA", (0, 0))[0]
    lds = cs.get("SQ_LDS_BANK_CONFLICT", (0, 0))[0]
    if not wc:
        continue
    print(f"{short}")
    print(f"   wait/wave={wait/wc:.2f} active/wave={act/wc:.2f} "
          f"conflict/active={lds/max(act,1):.3f} mfma_insts={mfma:.3g}")
