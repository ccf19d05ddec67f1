#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db: per-kernel time and (if present) PMC
counters. Usage: python tools/prof_summary.py <results.db> [out.txt]"""

import sqlite3
import sys


def main():
    path = sys.argv[1]
    out = open(sys.argv[2], "w") if len(sys.argv) > 2 else sys.stdout
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0]
    sfx = kd[len("rocpd_kernel_dispatch_"):]

    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY 1 ORDER BY 3 DESC LIMIT 30""").fetchall()
    tot = sum(r[2] for r in rows)
    print("== kernel time ==", file=out)
    for name, n, ms, avg in rows:
        print(f"{ms:9.3f} ms n={n:6d} avg={avg:8.2f} us  {name[:90]}",
              file=out)
    print(f"total {tot:.1f} ms", file=out)

    pmc = [t for t in tables if t.startswith("rocpd_pmc_event")]
    if pmc:
        try:
            q = f"""
            SELECT ks.display_name AS kname, pi.name AS counter,
                   SUM(pe.value) AS val, COUNT(DISTINCT pe.dispatch_id...)
            """
            # schema discovery
            cols = [r[1] for r in cur.execute(
                f"PRAGMA table_info(rocpd_pmc_event_{sfx})")]
            print(f"\npmc_event columns: {cols}", file=out)
            rows = cur.execute(f"""
                SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
                FROM rocpd_pmc_event_{sfx} pe
                JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
                JOIN rocpd_kernel_dispatch_{sfx} k
                     ON pe.event_id = k.event_id
                JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
                GROUP BY 1, 2 ORDER BY 1, 2""").fetchall()
            print("\n== PMC (summed over dispatches) ==", file=out)
            for kname, counter, val, n in rows:
                print(f"{kname[:60]:60s} {counter:22s} {val:18.0f} (n={n})",
                      file=out)
        except Exception as e:
            print(f"PMC join failed: {e}", file=out)
            # dump raw-ish
            try:
                rows = cur.execute(f"""
                    SELECT pi.name, SUM(pe.value)
                    FROM rocpd_pmc_event_{sfx} pe
                    JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
                    GROUP BY 1""").fetchall()
                for name, val in rows:
                    print(f"{name:24s} {val:18.0f}", file=out)
            except Exception as e2:
                print(f"fallback failed too: {e2}", file=out)


if __name__ == "__main__":
    main()
