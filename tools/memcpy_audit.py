#!/usr/bin/env python3
"""Memcpy + cast-kernel audit from a rocprofv3 hip-trace db: sizes and the
kernels dispatched immediately before each DtoD copy (caller attribution)."""
import glob
import sqlite3
import sys
from collections import Counter, defaultdict

path = sys.argv[1] if len(sys.argv) > 1 else glob.glob("/tmp/c12d/*.db")[0]
db = sqlite3.connect(path)
cur = db.cursor()
tabs = [r[0] for r in cur.execute("select name from sqlite_master where type='table'")]
T = lambda p: next(t for t in tabs if t.startswith(p))
mc = T("rocpd_memory_copy")
cols = [r[1] for r in cur.execute(f"PRAGMA table_info({mc})")]
print("memory_copy cols:", cols)
rows = list(cur.execute(f"select * from {mc}"))
print("copies:", len(rows))
iname = cols.index("name_id") if "name_id" in cols else None
isize = cols.index("size") if "size" in cols else None
istart = cols.index("start")
strs = {r[0]: r[1] for r in cur.execute(f"select id, string from {T('rocpd_string')}")} if any(
    t.startswith("rocpd_string") for t in tabs) else {}
agg = Counter()
szagg = defaultdict(int)
for r in rows:
    nm = strs.get(r[iname], str(r[iname])) if iname is not None else "?"
    sz = r[isize] if isize is not None else 0
    agg[(nm, sz)] += 1
    szagg[nm] += sz
for (nm, sz), n in agg.most_common(15):
    print(f"{n:6d} x {sz:>10} B  {nm}")
for nm, s in szagg.items():
    print(f"total {nm}: {s/1e6:.1f} MB")
# kernels right before big DtoD copies: correlate by start time
kd = T("rocpd_kernel_dispatch")
syms = {r[0]: r[1] for r in cur.execute(
    f"select id, display_name from {T('rocpd_info_kernel_symbol')}")}
disp = sorted(cur.execute(f"select start, kernel_id from {kd}"))
import bisect
starts = [d[0] for d in disp]
pred = Counter()
for r in rows:
    if isize is not None and r[isize] < 4096:
        continue
    i = bisect.bisect_left(starts, r[istart]) - 1
    if i >= 0:
        pred[syms.get(disp[i][1], "?")[:70]] += 1
print("\nkernel immediately before each copy (>=4KB):")
for k, n in pred.most_common(12):
    print(f"{n:6d}  {k}")
