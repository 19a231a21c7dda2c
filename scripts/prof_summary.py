#!/usr/bin/env python3
"""Summarize a rocprofv3 results db: total time per kernel symbol.

  python scripts/prof_summary.py out.db [topN]

Prints a markdown table (total ms, calls, avg us, kernel).  Used on the
GPU box so only the small digest travels back (the raw dbs exceed the
gpurun merge-back cap).
"""

import sqlite3
import sys


def main():
    path = sys.argv[1]
    topn = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "select name from sqlite_master where type='table'").fetchall()]
    pre = "rocpd_info_kernel_symbol_"
    sufs = [t.replace(pre, "") for t in tabs if t.startswith(pre)]
    rows = []
    for suf in sufs:
        rows += db.execute(f"""
            select ks.display_name, count(*), sum(kd.end-kd.start)/1e6
            from rocpd_kernel_dispatch_{suf} kd
            join rocpd_info_kernel_symbol_{suf} ks on ks.id = kd.kernel_id
            group by 1""").fetchall()
    agg = {}
    for name, n, ms in rows:
        nm = name.split("(")[0][:78]
        a = agg.setdefault(nm, [0, 0.0])
        a[0] += n
        a[1] += ms
    out = sorted(agg.items(), key=lambda kv: -kv[1][1])
    total = sum(v[1] for _, v in out)
    print(f"total kernel ms: {total:.1f}")
    print("| total ms | calls | avg us | kernel |")
    print("|---|---|---|---|")
    for nm, (n, ms) in out[:topn]:
        print(f"| {ms:.2f} | {n} | {ms / n * 1e3:.1f} | `{nm}` |")


if __name__ == "__main__":
    main()
