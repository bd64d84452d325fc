#!/usr/bin/env python3
"""GPU idle-gap analysis of a rocprofv3 kernel-trace .db (run on the box).

Uses adamw dispatches as step markers: for each inter-adamw window (one train
step) computes wall span, busy union and the largest idle gaps, with the
kernel names on each side of the gap. Answers "where does wall - kernel_busy
go?" for the graph-replayed train step."""
import re
import sqlite3
import sys


def load(db_path):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0]
    sfx = kd.replace("rocpd_kernel_dispatch_", "")
    rows = cur.execute(f"""
        SELECT kd.start, kd.end, ks.display_name
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        ORDER BY kd.start""").fetchall()
    return rows


def busy_union(rows):
    total, cur_s, cur_e = 0, None, None
    for s, e, _ in rows:
        if cur_e is None or s > cur_e:
            if cur_e is not None:
                total += cur_e - cur_s
            cur_s, cur_e = s, e
        else:
            cur_e = max(cur_e, e)
    if cur_e is not None:
        total += cur_e - cur_s
    return total


def analyze(db_path, out_path=None, marker="adamw", n_gaps=20):
    rows = load(db_path)
    raw = [i for i, (_, _, n) in enumerate(rows) if marker in n]
    # a step can fire the marker kernel more than once back-to-back (e.g. two
    # fused optimizer launches) — keep only the last of each cluster (<1ms)
    marks = [i for j, i in enumerate(raw)
             if j == len(raw) - 1 or rows[raw[j + 1]][0] - rows[i][1] > 1e6]
    lines = [f"dispatches={len(rows)} step_markers={len(marks)} (raw {len(raw)})"]
    lines.append("per-interval (marker->marker = one step incl. host work):")
    for j in range(max(0, len(marks) - 14), len(marks) - 1):
        win = rows[marks[j]:marks[j + 1]]
        span = win[-1][1] - win[0][0]
        busy = busy_union(win)
        gaps = []
        prev_e = win[0][1]
        for s, e, n in win[1:]:
            if s > prev_e:
                gaps.append(s - prev_e)
            prev_e = max(prev_e, e)
        big = sorted(gaps, reverse=True)[:3]
        lines.append(
            f"  step[{j}]: span={span/1e6:7.3f}ms busy={busy/1e6:7.3f}ms "
            f"idle={100*(span-busy)/span:4.1f}% ngaps={len(gaps)} "
            f"top_gaps_us={[round(g/1e3,1) for g in big]}")
    if len(marks) >= 3:
        # steady state: between the 2nd and last marker
        lo, hi = marks[1], marks[-1]
        win = rows[lo:hi]
        span = win[-1][1] - win[0][0]
        busy = busy_union(win)
        nsteps = len(marks) - 2
        lines.append(
            f"steady window: {nsteps} steps span={span/1e6:.2f}ms "
            f"busy={busy/1e6:.2f}ms idle={100*(span-busy)/span:.1f}% "
            f"({span/nsteps/1e6:.3f} ms/step wall, {busy/nsteps/1e6:.3f} busy)")
        gaps = []
        prev_e, prev_n = win[0][1], win[0][2]
        for s, e, n in win[1:]:
            if s > prev_e:
                gaps.append((s - prev_e, prev_n, n))
            if e > prev_e:
                prev_e, prev_n = e, n
        gaps.sort(reverse=True)
        lines.append(f"gap_count={len(gaps)} "
                     f"gap_total={sum(g[0] for g in gaps)/1e6:.2f}ms")
        lines.append(f"top {n_gaps} gaps (us, after -> before):")
        for d, a, b in gaps[:n_gaps]:
            a = re.sub(r"<[^>]*>", "", a)[:60]
            b = re.sub(r"<[^>]*>", "", b)[:60]
            lines.append(f"  {d/1e3:9.1f}  {a}  ->  {b}")
        # gap histogram
        buckets = [(2e3, "<2us"), (5e3, "2-5us"), (2e4, "5-20us"),
                   (1e5, "20-100us"), (1e18, ">100us")]
        hist = {lbl: [0, 0] for _, lbl in buckets}
        for d, _, _ in gaps:
            for thr, lbl in buckets:
                if d < thr:
                    hist[lbl][0] += 1
                    hist[lbl][1] += d
                    break
        lines.append("gap histogram: " + "  ".join(
            f"{lbl}:{c}({t/1e6:.2f}ms)" for lbl, (c, t) in hist.items()))
    text = "\n".join(lines) + "\n"
    if out_path:
        with open(out_path, "w") as f:
            f.write(text)
    return text


if __name__ == "__main__":
    out = sys.argv[2] if len(sys.argv) > 2 else None
    print(analyze(sys.argv[1], out))
