#!/usr/bin/env python3
"""Aggregate rocprofv3 --pmc counters per kernel from a results db."""
import re
import sqlite3
import sys
from collections import defaultdict


def summarize(db_path, out_path=None):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0].replace(
        "rocpd_kernel_dispatch_", "")
    # counter id -> name
    pmc_info = {r[0]: r[1] for r in cur.execute(
        f"SELECT id, name FROM rocpd_info_pmc_{sfx}")}
    rows = cur.execute(f"""
        SELECT ks.display_name, pe.pmc_id, SUM(pe.value), COUNT(DISTINCT pe.event_id)
        FROM rocpd_pmc_event_{sfx} pe
        JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id = kd.id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name, pe.pmc_id""").fetchall()
    agg = defaultdict(dict)
    calls = {}
    for name, pmc_id, val, cnt in rows:
        short = re.sub(r"<[^>]*>", "", name)[:60]
        agg[short][pmc_info.get(pmc_id, str(pmc_id))] = val
        calls[short] = cnt
    counters = sorted({c for v in agg.values() for c in v})
    lines = ["kernel | calls | " + " | ".join(counters)]
    for kname, vals in sorted(agg.items(), key=lambda kv: -max(kv[1].values())):
        lines.append(f"{kname} | {calls[kname]} | " +
                     " | ".join(f"{vals.get(c, 0):.3e}" for c in counters))
    text = "\n".join(lines) + "\n"
    if out_path:
        open(out_path, "w").write(text)
    return text


if __name__ == "__main__":
    print(summarize(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None))
