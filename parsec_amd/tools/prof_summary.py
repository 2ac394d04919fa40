#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db (kernel dispatches) into a markdown table.

Usage: python -m parsec_amd.tools.prof_summary results.db > summary.md
Mirrors the reference's dbp trace readers (tools/profiling/dbpreader.c) for
the rocprofv3 SQLite format.
"""
import sqlite3
import sys
from collections import Counter


def summarize(path, out=sys.stdout):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tabs if t.startswith('rocpd_kernel_dispatch_')][0]
    ks = [t for t in tabs if t.startswith('rocpd_info_kernel_symbol_')][0]
    rows = cur.execute(
        f"SELECT d.start,d.end,k.kernel_name FROM {kd} d "
        f"JOIN {ks} k ON d.kernel_id=k.id ORDER BY d.start").fetchall()
    if not rows:
        print("no kernel dispatches", file=out)
        return
    t0 = rows[0][0]
    t1 = max(r[1] for r in rows)
    merged = []
    for s, e, _ in rows:
        if merged and s <= merged[-1][1]:
            merged[-1][1] = max(merged[-1][1], e)
        else:
            merged.append([s, e])
    busy = sum(e - s for s, e in merged)
    wall = t1 - t0
    ktot = sum(e - s for s, e, _ in rows)
    agg, cnt = Counter(), Counter()
    for s, e, nm in rows:
        key = nm.split('(')[0][:60]
        agg[key] += e - s
        cnt[key] += 1
    print(f"# Kernel profile: {path}", file=out)
    print(f"\n- kernels: {len(rows)}", file=out)
    print(f"- wall (first..last kernel): {wall/1e6:.0f} ms", file=out)
    print(f"- GPU busy: {busy/1e6:.0f} ms ({100*busy/wall:.1f}%)", file=out)
    print(f"- sum of kernel times: {ktot/1e6:.0f} ms "
          f"(avg concurrency while busy {ktot/busy:.2f})", file=out)
    print("\n| total ms | calls | avg us | kernel |", file=out)
    print("|---:|---:|---:|---|", file=out)
    for k, v in agg.most_common(20):
        print(f"| {v/1e6:.1f} | {cnt[k]} | {v/cnt[k]/1e3:.1f} | `{k}` |",
              file=out)


if __name__ == "__main__":
    summarize(sys.argv[1])
