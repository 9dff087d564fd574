"""Summarise a rocprofv3 --pmc sqlite db: per-kernel counter totals and
per-wave rates.  Usage: python tools/pmc_extract.py <results.db> [filter]"""
import sqlite3
import sys
from collections import defaultdict


def main():
    path = sys.argv[1]
    filt = sys.argv[2] if len(sys.argv) > 2 else ""
    db = sqlite3.connect(path)
    cur = db.cursor()
    cur.execute("SELECT name FROM sqlite_master WHERE type='table' "
                "AND name LIKE 'rocpd_pmc_event%'")
    sfx = cur.fetchone()[0][len("rocpd_pmc_event_"):]

    q = f"""
    SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
    FROM rocpd_pmc_event_{sfx} pe
    JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
    JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id = kd.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name, pi.name
    """
    agg = defaultdict(dict)
    for kname, cname, total, n in cur.execute(q):
        if filt and filt not in kname:
            continue
        agg[kname][cname] = (total, n)
    for kname in sorted(agg):
        row = agg[kname]
        waves = row.get("SQ_WAVES", (0, 1))[0]
        short = kname.split("(")[0][:86]
        print(f"== {short}  dispatches={next(iter(row.values()))[1]}")
        for cname in sorted(row):
            tot, _ = row[cname]
            per = tot / waves if waves else 0.0
            print(f"   {cname:<26} total={tot:<16.0f} per_wave={per:.1f}")


if __name__ == "__main__":
    main()
