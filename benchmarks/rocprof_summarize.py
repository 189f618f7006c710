#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""Summarize a rocprofv3 SQLite results DB: per-kernel time and PMC
counter totals.  Usage: rocprof_summarize.py <db-or-glob> [label]"""
import glob
import sqlite3
import sys


def main():
    pat = sys.argv[1]
    label = sys.argv[2] if len(sys.argv) > 2 else pat
    dbs = sorted(glob.glob(pat, recursive=True))
    if not dbs:
        print(f"no db matches {pat}")
        return
    for db in dbs:
        con = sqlite3.connect(db)
        cur = con.cursor()
        tables = [r[0] for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        sfx = [t for t in tables if t.startswith("rocpd_metadata_")][0]
        sfx = sfx.split("rocpd_metadata_")[-1]
        print(f"== {label}: {db}")
        # kernel dispatch times
        try:
            q = f"""SELECT ks.display_name, COUNT(*),
                       SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e3
                    FROM rocpd_kernel_dispatch_{sfx} k
                    JOIN rocpd_info_kernel_symbol_{sfx} ks
                      ON k.kernel_id = ks.id
                    GROUP BY ks.display_name
                    ORDER BY 3 DESC LIMIT 12"""
            for name, n, tot, avg in cur.execute(q):
                print(f"  {tot:9.2f} ms n={n:5d} avg={avg:9.1f} us "
                      f"{name[:70]}")
        except sqlite3.Error as e:
            print("  (no dispatch table)", e)
        # PMC events joined to kernels
        try:
            cols = [c[1] for c in cur.execute(
                f"PRAGMA table_info(rocpd_pmc_event_{sfx})")]
            q = f"""SELECT ks.display_name, pi.name,
                           SUM(pe.value), COUNT(*)
                    FROM rocpd_pmc_event_{sfx} pe
                    JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
                    JOIN rocpd_kernel_dispatch_{sfx} k
                      ON pe.event_id = k.event_id
                    JOIN rocpd_info_kernel_symbol_{sfx} ks
                      ON k.kernel_id = ks.id
                    GROUP BY ks.display_name, pi.name
                    ORDER BY 3 DESC LIMIT 40"""
            rows = list(cur.execute(q))
            for name, cname, val, n in rows:
                print(f"  PMC {cname:24s} {val:16.0f} n={n:5d} "
                      f"{name[:52]}")
            if not rows:
                print("  (no joined pmc rows; dumping raw)")
                for r in cur.execute(
                        f"SELECT pmc_id, SUM(value), COUNT(*) FROM "
                        f"rocpd_pmc_event_{sfx} GROUP BY pmc_id LIMIT 10"):
                    print("   raw", r)
        except sqlite3.Error as e:
            print("  (pmc join failed)", e)


if __name__ == "__main__":
    main()
