#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results.db into a per-kernel counter table
(source of profiles/pmc_*.md). Joins rocpd_pmc_event.event_id ->
rocpd_kernel_dispatch.id -> rocpd_info_kernel_symbol, and pmc_id ->
rocpd_info_pmc. Derived columns when the standard counter set
(SQ_WAVE_CYCLES, SQ_WAIT_ANY, SQ_WAIT_INST_ANY, SQ_INSTS_MFMA) is
present: WAIT_ANY%% and WAIT_INST%% of wave cycles.

Usage: python tools/pmc_summary.py <glob-to-results.db> [min_gcycles]
"""
import glob
import re
import sqlite3
import sys


def main(dbglob, min_gcycles=0.5):
    db = sqlite3.connect(glob.glob(dbglob)[0])
    sfx = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_pmc_event%'"
    )][0].split("rocpd_pmc_event_")[1]
    counters = dict(db.execute(f"SELECT id, name FROM rocpd_info_pmc_{sfx}"))
    rows = db.execute(f"""
        SELECT s.display_name, e.pmc_id, SUM(e.value)
        FROM rocpd_pmc_event_{sfx} e
        JOIN rocpd_kernel_dispatch_{sfx} d ON d.id = e.event_id
        JOIN rocpd_info_kernel_symbol_{sfx} s ON s.id = d.kernel_id
        GROUP BY s.display_name, e.pmc_id""")
    agg = {}
    for name, pid, v in rows:
        nm = re.sub(r"\(.*", "", name)[:80]
        agg.setdefault(nm, {})[counters[pid]] = v
    cyc, wait, winst, mfma = ("SQ_WAVE_CYCLES", "SQ_WAIT_ANY",
                              "SQ_WAIT_INST_ANY", "SQ_INSTS_MFMA")
    std = all(c in next(iter(agg.values()), {}) for c in (cyc, wait, winst, mfma))
    if std:
        print(f"{'cycles(G)':>10} {'WAIT_ANY%':>10} {'WAIT_INST%':>11} "
              f"{'MFMA(M)':>9}  kernel")
        for nm, c in sorted(agg.items(), key=lambda kv: -kv[1].get(cyc, 0)):
            g = c.get(cyc, 0) / 1e9
            if g < min_gcycles:
                continue
            print(f"{g:>10.1f} {100 * c.get(wait, 0) / max(c.get(cyc, 1), 1):>10.1f} "
                  f"{100 * c.get(winst, 0) / max(c.get(cyc, 1), 1):>11.1f} "
                  f"{c.get(mfma, 0) / 1e6:>9.0f}  {nm}")
    else:  # arbitrary counter set: raw totals
        names = sorted({k for c in agg.values() for k in c})
        print("kernel\t" + "\t".join(names))
        for nm, c in sorted(agg.items(),
                            key=lambda kv: -max(kv[1].values(), default=0)):
            print(nm + "\t" + "\t".join(str(c.get(k, 0)) for k in names))


if __name__ == "__main__":
    main(sys.argv[1],
         float(sys.argv[2]) if len(sys.argv) > 2 else 0.5)
