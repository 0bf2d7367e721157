#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite DB: per-kernel dispatch stats and, for
PMC runs, per-kernel counter totals.  Used to produce profiles/*.txt from
gpurun rocprof captures."""
import sqlite3
import sys
from collections import defaultdict


def suffix(db):
    row = db.execute(
        "select name from sqlite_master where type='table' "
        "and name like 'rocpd_kernel_dispatch%'").fetchone()
    return row[0][len("rocpd_kernel_dispatch"):]


def kernel_names(db, sfx):
    names = {}
    for kid, name in db.execute(
            f"select id, display_name from rocpd_info_kernel_symbol{sfx}"):
        names[kid] = name
    return names


def main(path):
    db = sqlite3.connect(path)
    sfx = suffix(db)
    names = kernel_names(db, sfx)
    stats = defaultdict(lambda: [0, 0.0, float("inf"), 0.0])  # n, sum, min, max
    for kid, start, end in db.execute(
            f"select kernel_id, start, end from rocpd_kernel_dispatch{sfx}"):
        d = (end - start) / 1e3  # ns -> us
        s = stats[kid]
        s[0] += 1
        s[1] += d
        s[2] = min(s[2], d)
        s[3] = max(s[3], d)
    total = sum(s[1] for s in stats.values())
    print(f"{'kernel':<60} {'n':>6} {'total_us':>12} {'avg_us':>10} "
          f"{'min_us':>10} {'max_us':>10} {'pct':>6}")
    for kid, s in sorted(stats.items(), key=lambda kv: -kv[1][1]):
        nm = names.get(kid, str(kid)).split("(")[0][:58]
        print(f"{nm:<60} {s[0]:>6} {s[1]:>12.1f} {s[1]/s[0]:>10.2f} "
              f"{s[2]:>10.2f} {s[3]:>10.2f} {100*s[1]/total:>5.1f}%")

    # PMC counters, if present
    try:
        pmc_info = dict(db.execute(
            f"select id, name from rocpd_info_pmc{sfx}"))
    except sqlite3.OperationalError:
        pmc_info = {}
    if pmc_info:
        rows = db.execute(
            f"select e.pmc_id, d.kernel_id, e.value from rocpd_pmc_event{sfx} e "
            f"join rocpd_kernel_dispatch{sfx} d on e.event_id = d.event_id").fetchall()
        agg = defaultdict(lambda: [0, 0.0])
        for pid, kid, val in rows:
            a = agg[(pid, kid)]
            a[0] += 1
            a[1] += val
        print("\nPMC counters (value is counter units; FETCH_SIZE/WRITE_SIZE are KB "
              "per rocprof, and on gfx950 FETCH_SIZE reads 1/2 of wide coalesced "
              "streams - see MI355X_MICROARCH.md):")
        print(f"{'counter':<20} {'kernel':<50} {'n':>6} {'total':>16} {'avg':>14}")
        for (pid, kid), (n, tot) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
            nm = names.get(kid, str(kid)).split("(")[0][:48]
            print(f"{pmc_info.get(pid, pid):<20} {nm:<50} {n:>6} {tot:>16.0f} "
                  f"{tot/n:>14.1f}")


if __name__ == "__main__":
    for p in sys.argv[1:]:
        print(f"==== {p} ====")
        main(p)
