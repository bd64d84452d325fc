#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db into a small per-kernel table (run on
the GPU box so only the summary travels back, not the multi-MB sqlite)."""
import re
import sqlite3
import sys


def summarize(db_path, out_path=None, top=40):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0]
    sfx = kd.replace("rocpd_kernel_dispatch_", "")
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end - kd.start)/1e6,
               AVG(kd.end - kd.start)/1e3
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}""").fetchall()
    total = cur.execute(
        f"SELECT SUM(end-start)/1e6, COUNT(*) FROM rocpd_kernel_dispatch_{sfx}"
    ).fetchone()
    lines = [f"total_kernel_ms={total[0]:.2f} dispatches={total[1]}",
             f"{'kernel':<84} {'calls':>7} {'ms':>9} {'avg_us':>8} {'pct':>5}"]
    for name, cnt, ms, avg in rows:
        n = re.sub(r"<[^>]*>", "", name)[:82]
        lines.append(f"{n:<84} {cnt:>7} {ms:>9.2f} {avg:>8.1f} "
                     f"{100*ms/total[0]:>5.1f}")
    text = "\n".join(lines) + "\n"
    if out_path:
        with open(out_path, "w") as f:
            f.write(text)
    return text


if __name__ == "__main__":
    out = sys.argv[2] if len(sys.argv) > 2 else None
    top = int(sys.argv[3]) if len(sys.argv) > 3 else 40
    print(summarize(sys.argv[1], out, top))
