#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results.db: per-kernel counter totals and
derived ratios (VALU issue share, MFMA share, LDS conflicts, wait share).
"""
import collections
import re
import sqlite3
import sys


def main():
    db = sys.argv[1]
    con = sqlite3.connect(db)
    cur = con.cursor()
    sfx = None
    for (name,) in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        m = re.match(r"rocpd_pmc_event_(.*)", name)
        if m:
            sfx = m.group(1)
    pm = {i: n for i, n in
          cur.execute(f"SELECT id, name FROM rocpd_info_pmc_{sfx}")}
    kd = {e: (n, dur) for e, n, dur in cur.execute(f"""
        SELECT d.event_id, k.display_name, d.end - d.start
        FROM rocpd_kernel_dispatch_{sfx} d
        JOIN rocpd_info_kernel_symbol_{sfx} k ON d.kernel_id = k.id""")}

    def short(n):
        m = re.search(r"(\w+_kernel)", n)
        return m.group(1) if m else re.sub(r"<.*", "", n).split("(")[0][-40:]

    agg = collections.defaultdict(lambda: collections.Counter())
    dur = collections.Counter()
    cnt = collections.Counter()
    for row in cur.execute(f"SELECT event_id, pmc_id, value "
                           f"FROM rocpd_pmc_event_{sfx}"):
        eid, pid, val = row
        if eid not in kd:
            continue
        kn = short(kd[eid][0])
        agg[kn][pm.get(pid, pid)] += val
    for e, (n, d) in kd.items():
        kn = short(n)
        dur[kn] += d
        cnt[kn] += 1
    print(f"{'kernel':42s} {'calls':>6s} {'ms':>8s} {'VALU%':>6s} "
          f"{'MFMA%':>6s} {'wait%':>6s} {'ldsconf%':>8s}")
    rows = sorted(agg.items(), key=lambda kv: -dur[kv[0]])
    for kn, c in rows:
        wc = c.get("SQ_WAVE_CYCLES", 0) or 1
        # VALU issue cost: 4 cycles per wave inst on a 16-wide SIMD
        valu = 400.0 * c.get("SQ_INSTS_VALU", 0) * 4 / wc
        mfma = 400.0 * c.get("SQ_INSTS_MFMA", 0) * 4 / wc
        wait = 100.0 * c.get("SQ_WAIT_ANY", 0) / wc
        lds = 100.0 * c.get("SQ_LDS_BANK_CONFLICT", 0) / wc
        print(f"{kn:42s} {cnt[kn]:6d} {dur[kn]/1e6:8.2f} {valu/4:6.1f} "
              f"{mfma/4:6.1f} {wait:6.1f} {lds:8.2f}")


if __name__ == "__main__":
    main()
