#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc events per kernel from a results db."""
import glob
import sqlite3
import sys


def main(path):
    dbs = glob.glob(path) if "*" in path else [path]
    c = sqlite3.connect(dbs[0]).cursor()
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [t for t in tabs if t.startswith("rocpd_pmc_event")][0].replace(
        "rocpd_pmc_event_", "")
    rows = c.execute(f"""
        SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
        FROM rocpd_pmc_event_{sfx} pe
        JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
        JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.dispatch_id = kd.dispatch_id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name, pi.name ORDER BY 1, 2""").fetchall()
    cur = None
    for name, ctr, val, n in rows:
        short = name.split("(")[0][:70]
        if short != cur:
            print(f"\n{short}  ({n} dispatches)")
            cur = short
        print(f"  {ctr:<28} {val:>18.0f}")


if __name__ == "__main__":
    main(sys.argv[1])
