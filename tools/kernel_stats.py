#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into per-kernel totals (ms, count)."""
import collections
import re
import sqlite3
import sys


def load(db_path):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    sfx = None
    for (name,) in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        m = re.match(r"rocpd_kernel_dispatch_(.*)", name)
        if m:
            sfx = m.group(1)
    assert sfx, "no kernel dispatch table"
    rows = list(cur.execute(f"""
        SELECT k.display_name, COUNT(*), SUM(d.end - d.start)
        FROM rocpd_kernel_dispatch_{sfx} d
        JOIN rocpd_info_kernel_symbol_{sfx} k ON d.kernel_id = k.id
        GROUP BY k.display_name"""))
    return rows


def short(n):
    m = re.search(r"(\w+_kernel)", n)
    if m:
        return m.group(1)
    n2 = re.sub(r"<.*", "", n).split("(")[0]
    return (n2.split("::")[-1] or n[:50])[:55]


def main():
    db = sys.argv[1]
    steps = float(sys.argv[2]) if len(sys.argv) > 2 else 10.0
    agg = collections.Counter()
    cnt = collections.Counter()
    total = 0
    for name, c, dur in load(db):
        agg[short(name)] += dur
        cnt[short(name)] += c
        total += dur
    print(f"total GPU kernel time: {total/1e6:.1f} ms "
          f"-> {total/1e6/steps:.2f} ms/step over {steps:.0f} steps")
    print(f"{'kernel':45s} {'count':>7s} {'/step':>6s} {'ms/step':>8s} "
          f"{'us/call':>8s} {'%':>5s}")
    for name, dur in agg.most_common(45):
        print(f"{name:45s} {cnt[name]:7d} {cnt[name]/steps:6.0f} "
              f"{dur/1e6/steps:8.3f} {dur/1e3/cnt[name]:8.1f} "
              f"{100*dur/total:5.1f}")


if __name__ == "__main__":
    main()
