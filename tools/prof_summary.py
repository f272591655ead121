#!/usr/bin/env python3
"""Summarize a rocprofv3 SQLite result into a small per-kernel table.

Usage: python tools/prof_summary.py results.db [--window-ms N] [-o out.txt]
Groups kernel dispatches by name over the last N ms of the trace (default:
whole trace) and prints time share, counts and mean duration.
"""
import argparse
import sqlite3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--window-ms", type=float, default=0.0,
                    help="only the last N ms of the trace")
    ap.add_argument("--skip-ms", type=float, default=0.0,
                    help="drop the first N ms (warmup/find)")
    ap.add_argument("-o", "--out", default=None)
    ap.add_argument("--limit", type=int, default=40)
    args = ap.parse_args()

    db = sqlite3.connect(args.db)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE 'rocpd_kernel_dispatch%'")]
    sfx = tabs[0].replace("rocpd_kernel_dispatch_", "")
    lo, hi = cur.execute(f"SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_{sfx}").fetchone()
    t0 = lo + args.skip_ms * 1e6
    if args.window_ms > 0:
        t0 = max(t0, hi - args.window_ms * 1e6)
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} k
    JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id=s.id
    WHERE k.start >= {t0}
    GROUP BY s.display_name ORDER BY 3 DESC
    """
    rows = cur.execute(q).fetchall()
    span_ms = (hi - t0) / 1e6
    out = []
    tot = sum(r[2] for r in rows)
    totn = sum(r[1] for r in rows)
    out.append(f"window: {span_ms:.1f} ms   kernel time: {tot:.1f} ms "
               f"({100*tot/span_ms:.0f}% busy)   dispatches: {totn}")
    out.append(f"{'ms':>9} {'%':>5} {'n':>6} {'avg_us':>8}  kernel")
    for name, cnt, ms, us in rows[:args.limit]:
        out.append(f"{ms:9.2f} {100*ms/tot:5.1f} {cnt:6d} {us:8.1f}  {name[:100]}")
    text = "\n".join(out)
    if args.out:
        with open(args.out, "w") as f:
            f.write(text + "\n")
    print(text)


if __name__ == "__main__":
    main()
