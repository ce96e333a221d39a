#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc counters_collection by kernel (defensive)."""
import glob
import sqlite3
import sys
import traceback

try:
    db = sqlite3.connect(sorted(glob.glob(sys.argv[1]))[-1])
    cols = [d[1] for d in db.execute(
        "PRAGMA table_info(counters_collection)")]
    print("# cols:", cols)
    name_col = next((c for c in cols if "kernel" in c.lower()
                     and "name" in c.lower()), None)
    cnt_cands = [c for c in cols if "counter" in c.lower()
                 or c.lower() in ("name", "symbol")]
    val_col = "value" if "value" in cols else next(
        (c for c in cols if "value" in c.lower()), None)
    print("# picked:", name_col, cnt_cands, val_col)
    if name_col is None:
        # fall back: find a text column containing kernel-looking strings
        for c in cols:
            try:
                v = db.execute(
                    f"SELECT {c} FROM counters_collection LIMIT 200"
                ).fetchall()
                if any(isinstance(x[0], str) and "kernel" in x[0]
                       for x in v):
                    name_col = c
                    break
            except Exception:
                pass
    cnt_col = None
    for c in cnt_cands:
        v = [x[0] for x in db.execute(
            f"SELECT DISTINCT {c} FROM counters_collection LIMIT 20")]
        if any(isinstance(x, str) and x.startswith("SQ_") for x in v):
            cnt_col = c
            break
    print("# final:", name_col, cnt_col, val_col)
    q = (f"SELECT {name_col}, {cnt_col}, SUM({val_col}) "
         f"FROM counters_collection GROUP BY {name_col}, {cnt_col}")
    agg = {}
    for kn, cn, v in db.execute(q):
        agg.setdefault(str(kn), {})[str(cn)] = v
    pats = sys.argv[2].split(",") if len(sys.argv) > 2 else None
    for kn, cs in sorted(agg.items()):
        if pats and not any(p in kn for p in pats):
            continue
        wc = cs.get("SQ_WAVE_CYCLES", 0)
        if not wc:
            continue
        wait = cs.get("SQ_WAIT_ANY", 0)
        act = cs.get("SQ_ACTIVE_INST_ANY", 0)
        mfma = cs.get("SQ_INSTS_MFMA", 0)
        lds = cs.get("SQ_LDS_BANK_CONFLICT", 0)
        short = kn.replace('(anonymous namespace)::', '')
        print(short.split('(')[0][:58] or short[:58])
        print(f"   wait/wave={wait/wc:.2f} active/wave={act/wc:.2f} "
              f"conflict/active={lds/max(act,1):.3f} "
              f"mfma_insts={mfma:.3g}")
except Exception:
    traceback.print_exc()
