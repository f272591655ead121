#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc output db: per-kernel counter aggregates.

Usage: python tools/pmc_summary.py results.db [-o out.txt]
"""
import argparse
import sqlite3
from collections import defaultdict


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("-o", "--out", default=None)
    args = ap.parse_args()
    db = sqlite3.connect(args.db)
    cur = db.cursor()

    def tab(prefix):
        r = [x[0] for x in cur.execute(
            "SELECT name FROM sqlite_master WHERE name LIKE ? AND name != ?",
            (prefix + "_%", prefix))]
        return r[0] if r else prefix

    pmc_t = tab("rocpd_pmc_event")
    info_t = tab("rocpd_info_pmc")
    disp_t = tab("rocpd_kernel_dispatch")
    sym_t = tab("rocpd_info_kernel_symbol")

    names = {}
    try:
        cols = [c[1] for c in cur.execute(f"PRAGMA table_info({info_t})")]
        idcol = "id"
        namecol = "name" if "name" in cols else cols[1]
        for row in cur.execute(f"SELECT {idcol}, {namecol} FROM {info_t}"):
            names[row[0]] = row[1]
    except Exception as e:
        print("info table issue:", e)

    cols = [c[1] for c in cur.execute(f"PRAGMA table_info({pmc_t})")]
    print("pmc cols:", cols)
    # typical: id, guid, pmc_id(->info), event_id(->dispatch), value, extdata
    q = f"""
    SELECT p.pmc_id, s.display_name, SUM(p.value), COUNT(*)
    FROM {pmc_t} p
    JOIN {disp_t} k ON p.event_id = k.event_id
    JOIN {sym_t} s ON k.kernel_id = s.id
    GROUP BY p.pmc_id, s.display_name
    """
    out = []
    try:
        for pmc_id, kname, total, cnt in cur.execute(q):
            out.append(f"{names.get(pmc_id, pmc_id)!s:>28} {total:16.0f} (n={cnt})  {kname[:70]}")
    except Exception as e:
        out.append(f"join failed ({e}); raw aggregate by pmc id:")
        agg = defaultdict(float)
        cnt = defaultdict(int)
        for row in cur.execute(f"SELECT * FROM {pmc_t}"):
            # id, guid, pmc_id, event_id, value, ...
            agg[row[2]] += row[4]
            cnt[row[2]] += 1
        for k, v in agg.items():
            out.append(f"{names.get(k, k)!s:>28} {v:16.0f} (n={cnt[k]})")
    text = "\n".join(out)
    if args.out:
        open(args.out, "w").write(text + "\n")
    print(text)


if __name__ == "__main__":
    main()
