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
    # discover the join columns (schema varies across rocprofv3 builds)
    pe_cols = [r[1] for r in c.execute(f"PRAGMA table_info(rocpd_pmc_event_{sfx})")]
    print("# pmc_event columns:", pe_cols)
    kd_cols = [r[1] for r in c.execute(
        f"PRAGMA table_info(rocpd_kernel_dispatch_{sfx})")]
    print("# kernel_dispatch columns:", kd_cols)
    link = [col for col in pe_cols if "dispatch" in col or "event" in col or "id" in col]
    # common layout: pmc_event(event_id|dispatch_row, pmc_id, value); try via
    # event_id -> kernel_dispatch.id
    join_col = "event_id" if "event_id" in pe_cols else link[0]
    rows = c.execute(f"""
        SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
        FROM rocpd_pmc_event_{sfx} pe
        JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
        JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.{join_col} = kd.id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name, pi.name ORDER BY 1, 2""").fetchall()
    import re
    cur = None
    for name, ctr, val, n in rows:
        short = re.sub(r"\(anonymous namespace\)::", "", name)
        short = re.sub(r"^void ", "", short).split("(")[0][:70]
        if short != cur:
            print(f"\n{short}  ({n} dispatches)")
            cur = short
        print(f"  {ctr:<28} {val:>18.0f}")


if __name__ == "__main__":
    main(sys.argv[1])
