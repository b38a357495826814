#!/usr/bin/env python3
"""Per-kernel waves-in-flight from a rocprofv3 --pmc results.db.

Needs SQ_WAVE_CYCLES + SQ_BUSY_CU_CYCLES (and optionally SQ_WAIT_ANY,
SQ_INSTS_MFMA).  waves-in-flight/SIMD = WAVE_CYCLES / BUSY_CU_CYCLES / 4
answers "is this kernel occupancy-limited" directly (docs/ROADMAP.md §5).
"""
import collections
import re
import sqlite3
import sys


def main():
    con = sqlite3.connect(sys.argv[1])
    cur = con.cursor()
    sfx = None
    for (name,) in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        m = re.match(r"rocpd_pmc_event_(.*)", name)
        if m:
            sfx = m.group(1)
    pm = {i: n for i, n in
          cur.execute(f"SELECT id, name FROM rocpd_info_pmc_{sfx}")}
    kd = {e: n for e, n in cur.execute(f"""
        SELECT d.event_id, k.display_name
        FROM rocpd_kernel_dispatch_{sfx} d
        JOIN rocpd_info_kernel_symbol_{sfx} k ON d.kernel_id = k.id""")}
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    for e, pid, val in cur.execute(
            f"SELECT event_id, pmc_id, value FROM rocpd_pmc_event_{sfx}"):
        if e in kd:
            kn = re.search(r"(\w+_kernel)", kd[e])
            agg[kn.group(1) if kn else kd[e][:30]][pm[pid]] += val
    for k, d in sorted(agg.items()):
        wc = d.get("SQ_WAVE_CYCLES", 0)
        busy = d.get("SQ_BUSY_CU_CYCLES", 0)
        wait = d.get("SQ_WAIT_ANY", 0)
        mfma = d.get("SQ_INSTS_MFMA", 0)
        if busy:
            print(f"{k:28s} waves/SIMD={wc / busy / 4:5.2f}  "
                  f"wait%={100 * wait / max(wc, 1):5.1f}  mfma={mfma:.2e}")


if __name__ == "__main__":
    main()
