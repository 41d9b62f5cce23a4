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

# --- copy/cast KERNELS (rocclr copyBuffer, at::copy casts): predecessors ---
targets = ("copyBuffer", "bfloat16tofloat32", "bfloat16_copy", "float32tobfloat16")
full = sorted(cur.execute(f"select start, kernel_id from {kd}"))
names = [syms.get(k, "?") for _, k in full]
pred2 = Counter()
tot = Counter()
for i, nm in enumerate(names):
    for t in targets:
        if t in nm:
            tot[t] += 1
            for j in range(i - 1, max(-1, i - 4), -1):
                if all(x not in names[j] for x in targets):
                    pred2[(t, names[j][:64])] += 1
                    break
            break
print("\ncopy/cast kernel counts:", dict(tot))
for (t, k), n in pred2.most_common(16):
    print(f"{n:6d}  {t:22s} after {k}")

# --- temporal: are copy/cast kernels setup-time or steady-state? ---
all_start = full[0][0]
all_end = max(s for s, _ in full)
span = all_end - all_start
buckets = Counter()
for i, nm in enumerate(names):
    for t in targets:
        if t in nm:
            frac = (full[i][0] - all_start) / span
            buckets[(t, min(9, int(frac * 10)))] += 1
            break
print("\ntemporal deciles (0=start .. 9=end):")
for t in targets:
    row = [buckets.get((t, d), 0) for d in range(10)]
    if sum(row):
        print(f"{t:22s} {row}")
