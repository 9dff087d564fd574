"""Summarise a rocprofv3 --kernel-trace sqlite db as a per-kernel time
table (the format committed under profiles/).

Usage: python tools/kstats_extract.py <results.db> [min_pct]
"""
import sqlite3
import sys


def main():
    path = sys.argv[1]
    min_pct = float(sys.argv[2]) if len(sys.argv) > 2 else 0.3
    db = sqlite3.connect(path)
    cur = db.cursor()
    cur.execute("SELECT name FROM sqlite_master WHERE type='table' "
                "AND name LIKE 'rocpd_kernel_dispatch%'")
    sfx = cur.fetchone()[0][len("rocpd_kernel_dispatch_"):]
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1000.0
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC
    """
    rows = cur.execute(q).fetchall()
    tot = sum(r[2] for r in rows)
    print(f"total kernel time: {tot / 1000:.1f} ms")
    print("| kernel | calls | total us | avg us | % |")
    print("|---|---|---|---|---|")
    for name, calls, us in rows:
        pct = 100 * us / tot
        if pct < min_pct:
            continue
        short = name.split("(")[0].replace("void ", "")[:86]
        print(f"| `{short}` | {calls} | {us:.0f} | {us / calls:.1f} "
              f"| {pct:.2f} |")


if __name__ == "__main__":
    main()
