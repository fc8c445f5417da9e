#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a per-kernel us/step table
(replacement for the --stats stdout table; also computes the last-1s
window like profiles/final_step_profile.md)."""
import glob
import re
import sqlite3
import sys


def main(dbglob, steps=None):
    db = glob.glob(dbglob)[0]
    c = sqlite3.connect(db)
    sfx = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")][0].split("rocpd_kernel_dispatch_")[1]
    rows = list(c.execute(f"""
        SELECT s.display_name, d.start, d.end
        FROM rocpd_kernel_dispatch_{sfx} d
        JOIN rocpd_info_kernel_symbol_{sfx} s ON s.id = d.kernel_id"""))
    if not rows:
        print("no dispatches"); return
    tmax = max(r[2] for r in rows)
    w0 = tmax - 1_000_000_000           # last 1 s window
    agg = {}
    for name, st, en in rows:
        if en < w0:
            continue
        nm = re.sub(r"\(.*", "", name)[:90]
        a = agg.setdefault(nm, [0.0, 0])
        a[0] += (en - st) / 1e3
        a[1] += 1
    total = sum(v[0] for v in agg.values())
    # estimate steps in window from the once-per-step sgd kernel
    per_step = None
    for nm, v in agg.items():
        if "sgd_step" in nm or "adam_step" in nm:
            per_step = v[1]
    n = steps or per_step or 1
    print(f"# busy {total/n:.0f} us/step over ~{n} steps in last-1s window")
    print(f"{'us/step':>9} {'calls/step':>10}  kernel")
    for nm, (us, cnt) in sorted(agg.items(), key=lambda kv: -kv[1][0]):
        print(f"{us/n:9.1f} {cnt/n:10.1f}  {nm}")


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/prof5/runc/*.db",
         int(sys.argv[2]) if len(sys.argv) > 2 else None)
