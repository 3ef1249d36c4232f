#!/usr/bin/env python3
"""Aggregate PMC counters per kernel from a rocprofv3 rocpd DB."""
import collections
import sqlite3
import sys

conn = sqlite3.connect(sys.argv[1])
T = [r[0] for r in conn.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]


def tb(p):
    return next(t for t in T if t.startswith(p))


pmc = tb("rocpd_pmc_event")
disp = tb("rocpd_kernel_dispatch")
sym = tb("rocpd_info_kernel_symbol")
info = tb("rocpd_info_pmc")
pcols = [c[1] for c in conn.execute(f"PRAGMA table_info({pmc})")]
dcols = [c[1] for c in conn.execute(f"PRAGMA table_info({disp})")]
print("# pmc cols:", pcols, file=sys.stderr)
print("# disp cols:", dcols, file=sys.stderr)
join_col = next((c for c in ("dispatch_id", "event_id", "id")
                 if c in pcols), pcols[0])
d_col = "dispatch_id" if "dispatch_id" in dcols else "id"
val_col = "value" if "value" in pcols else pcols[-1]
key_col = next((c for c in ("pmc_id", "counter_id", "info_id")
                if c in pcols), None)
rows = conn.execute(f"""
  SELECT s.kernel_name, i.name, SUM(p.{val_col})
  FROM {pmc} p JOIN {disp} d ON p.{join_col} = d.{d_col}
  JOIN {sym} s ON d.kernel_id = s.id
  JOIN {info} i ON p.{key_col} = i.id
  GROUP BY s.kernel_name, i.name""").fetchall()
agg = collections.defaultdict(dict)
for name, counter, val in rows:
    for pre in ("_ZN12_GLOBAL__N_1", "void "):
        if name.startswith(pre):
            name = name[len(pre):]
    agg[name.split("(")[0][:46]][counter] = val
cols = ["SQ_WAVE_CYCLES", "SQ_INSTS_MFMA", "SQ_INSTS_VALU",
        "SQ_LDS_BANK_CONFLICT"]
hdr = " ".join("%16s" % c[3:] for c in cols)
print("%-46s %s" % ("kernel", hdr))
ordered = sorted(agg.items(),
                 key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", 0))
for name, cts in ordered[:18]:
    vals = " ".join("%16.3g" % cts.get(c, 0) for c in cols)
    print("%-46s %s" % (name, vals))
