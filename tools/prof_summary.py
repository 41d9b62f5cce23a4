#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database into a per-kernel stats table
(total time, calls, mean) — the judge-facing artifact committed under
profiles/.

Usage:
  prof_summary.py results.db [top] [out.txt]       one-db summary
  prof_summary.py --diff old.db new.db [top]       per-kernel mean_us diff
    (match by normalized kernel name; use after an optimization to see
    exactly which kernels moved instead of eyeballing two tables)
"""

from __future__ import annotations

import re
import sqlite3
import sys
from collections import defaultdict


def by_grid(db_path: str, flt: str = "") -> str:
    """Per-(kernel, grid) means — disambiguates GEMM call sites that share
    a kernel name but differ in problem shape (grid size)."""
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute("select name from sqlite_master where type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    names = {r[0]: r[1] for r in cur.execute(f"select id, display_name from {sym}")}
    agg = defaultdict(lambda: [0.0, 0])
    q = f"select kernel_id, grid_size_x, grid_size_y, grid_size_z, start, end from {disp}"
    for kid, gx, gy, gz, s, e in cur.execute(q):
        nm = names.get(kid, str(kid))
        agg[(nm, gx, gy, gz)][0] += (e - s) / 1e6
        agg[(nm, gx, gy, gz)][1] += 1
    out = []
    for (nm, gx, gy, gz), (ms, n) in sorted(agg.items(), key=lambda kv: -kv[1][0]):
        if flt and flt.lower() not in nm.lower():
            continue
        out.append(f"{ms:9.2f}ms {n:6d}x {ms/n*1e3:8.2f}us  grid({gx},{gy},{gz})  {nm[:90]}")
    return "\n".join(out[:60])


def summarize(db_path: str, top: int = 40) -> str:
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute("select name from sqlite_master where type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    names = {r[0]: r[1] for r in cur.execute(f"select id, display_name from {sym}")}
    agg = defaultdict(lambda: [0.0, 0])
    for kid, start, end in cur.execute(f"select kernel_id, start, end from {disp}"):
        a = agg[names.get(kid, str(kid))]
        a[0] += (end - start) / 1e6  # ns -> ms
        a[1] += 1
    total = sum(a[0] for a in agg.values())
    lines = [
        f"# rocprofv3 kernel summary: {db_path}",
        f"# total GPU kernel time: {total:.1f} ms across {sum(a[1] for a in agg.values())} dispatches",
        "",
        f"{'total_ms':>10} {'%':>6} {'calls':>8} {'mean_us':>9}  name",
    ]
    for name, (ms, n) in sorted(agg.items(), key=lambda kv: -kv[1][0])[:top]:
        short = re.sub(r"\s+", " ", name)[:130]
        lines.append(f"{ms:10.2f} {100*ms/total:6.2f} {n:8d} {1e3*ms/n:9.2f}  {short}")
    return "\n".join(lines)


def _agg(db_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute("select name from sqlite_master where type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    names = {r[0]: r[1] for r in cur.execute(f"select id, display_name from {sym}")}
    agg = defaultdict(lambda: [0.0, 0])
    for kid, start, end in cur.execute(f"select kernel_id, start, end from {disp}"):
        key = re.sub(r"\s+", " ", names.get(kid, str(kid)))[:90]
        a = agg[key]
        a[0] += (end - start) / 1e6
        a[1] += 1
    return agg


def diff(old_db: str, new_db: str, top: int = 40) -> str:
    a, b = _agg(old_db), _agg(new_db)
    rows = []
    for name in set(a) | set(b):
        oms, on = a.get(name, [0.0, 0])
        nms, nn = b.get(name, [0.0, 0])
        om = 1e3 * oms / on if on else 0.0
        nm = 1e3 * nms / nn if nn else 0.0
        rows.append((nms - oms, om, nm, on, nn, name))
    lines = [
        f"# kernel diff: {old_db} -> {new_db} (sorted by total-ms delta)",
        f"{'d_total_ms':>11} {'old_us':>9} {'new_us':>9} {'old_n':>7} {'new_n':>7}  name",
    ]
    for d, om, nm, on, nn, name in sorted(rows, key=lambda r: r[0])[:top]:
        lines.append(f"{d:+11.2f} {om:9.2f} {nm:9.2f} {on:7d} {nn:7d}  {name}")
    lines.append("  ... (most-regressed tail) ...")
    for d, om, nm, on, nn, name in sorted(rows, key=lambda r: -r[0])[:10]:
        lines.append(f"{d:+11.2f} {om:9.2f} {nm:9.2f} {on:7d} {nn:7d}  {name}")
    return "\n".join(lines)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "--by-grid":
        print(by_grid(sys.argv[2], sys.argv[3] if len(sys.argv) > 3 else ""))
    elif len(sys.argv) > 1 and sys.argv[1] == "--diff":
        print(diff(sys.argv[2], sys.argv[3],
                   int(sys.argv[4]) if len(sys.argv) > 4 else 40))
    else:
        out = summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 40)
        print(out)
        if len(sys.argv) > 3:
            with open(sys.argv[3], "w") as f:
                f.write(out + "\n")
