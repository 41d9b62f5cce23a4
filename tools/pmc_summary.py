#!/usr/bin/env python3
"""Per-kernel PMC means from a rocprofv3 rocpd SQLite db.

Usage: pmc_summary.py results.db [name_filter]
"""

import sqlite3
import sys
from collections import defaultdict


def main():
    path = sys.argv[1]
    flt = sys.argv[2] if len(sys.argv) > 2 else ""
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute("select name from sqlite_master where type='table'")]

    def tab(prefix):
        return next(t for t in tabs if t.startswith(prefix))

    pmc_info = tab("rocpd_info_pmc")
    pmc_ev = tab("rocpd_pmc_event")
    disp = tab("rocpd_kernel_dispatch")
    sym = tab("rocpd_info_kernel_symbol")
    names = {r[0]: r[1] for r in cur.execute(f"select id, display_name from {sym}")}
    pmc_names = {r[0]: r[1] for r in cur.execute(f"select id, name from {pmc_info}")}
    # dispatch: id, kernel_id, start, end (col names vary; fetch by name)
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]
    kid_col = "kernel_id"
    # pmc_event.event_id references the dispatch's event id
    ev_col = next((c for c in cols if c in ("event_id", "id")), "id")
    agg = defaultdict(lambda: defaultdict(lambda: [0.0, 0]))
    q = (f"select d.{kid_col}, p.pmc_id, p.value, d.start, d.end "
         f"from {pmc_ev} p join {disp} d on p.event_id = d.{ev_col}")
    times = defaultdict(lambda: [0.0, 0])
    for kid, pid, val, start, end in cur.execute(q):
        kname = names.get(kid, str(kid))
        a = agg[kname][pmc_names.get(pid, str(pid))]
        a[0] += val
        a[1] += 1
        t = times[kname]
        t[0] += (end - start) / 1e3
        t[1] += 1
    npmc = max(1, len(pmc_names))
    for kname, counters in sorted(agg.items()):
        if flt and flt not in kname:
            continue
        t = times[kname]
        print(f"{kname[:100]}  [mean {t[0]/max(t[1],1)*npmc:.1f} us x {t[1]//npmc}]")
        vals = {}
        for cname, (s, n) in sorted(counters.items()):
            vals[cname] = s / n
            print(f"   {cname:<30} {s/n:12.4g}")
        wc = vals.get("SQ_WAVE_CYCLES")
        mf = vals.get("SQ_VALU_MFMA_BUSY_CYCLES")
        wa = vals.get("SQ_WAIT_ANY")
        if wc:
            line = "  "
            if mf is not None:
                line += f" mfma_busy_fraction {mf/wc:.3f}"
            if wa is not None:
                line += f" wait_fraction {wa/wc:.3f}"
            print(line)
        print()


if __name__ == "__main__":
    main()
