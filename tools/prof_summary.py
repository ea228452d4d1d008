#!/usr/bin/env python3
"""Summarize a rocprofv3 results db: total time per kernel name.

Usage: python tools/prof_summary.py gpurun_out/prof/bench1_results.db [N]
"""
import sqlite3
import sys


def summarize(path: str, top: int = 30):
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")]
    ks = [t for t in tabs if t.startswith("rocpd_info_kernel_symbol")]
    st = [t for t in tabs if t.startswith("rocpd_string")]
    if not kd:
        print("no kernel dispatch table")
        return
    rows = []
    for t, s, strt in zip(kd, ks, st):
        q = f"""
        SELECT sym.display_name, COUNT(*), SUM(d.end - d.start),
               AVG(d.end - d.start)
        FROM {t} d JOIN {s} sym ON d.kernel_id = sym.id
        GROUP BY sym.display_name
        """
        try:
            rows += db.execute(q).fetchall()
        except Exception as e:
            # fall back: resolve display_name via string table
            q2 = f"""
            SELECT str.string, COUNT(*), SUM(d.end - d.start),
                   AVG(d.end - d.start)
            FROM {t} d JOIN {s} sym ON d.kernel_id = sym.id
            JOIN {strt} str ON sym.display_name = str.id
            GROUP BY str.string
            """
            rows += db.execute(q2).fetchall()
    agg = {}
    for name, cnt, tot, avg in rows:
        name = str(name)
        if name in agg:
            c, t0 = agg[name]
            agg[name] = (c + cnt, t0 + tot)
        else:
            agg[name] = (cnt, tot)
    total = sum(t for _, t in agg.values())
    print(f"total kernel time: {total/1e6:.2f} ms over "
          f"{sum(c for c, _ in agg.values())} dispatches")
    print(f"{'time_ms':>10} {'pct':>6} {'count':>7} {'avg_us':>9}  name")
    for name, (cnt, tot) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:top]:
        print(f"{tot/1e6:10.2f} {100*tot/total:6.2f} {cnt:7d} "
              f"{tot/cnt/1e3:9.2f}  {name[:110]}")


if __name__ == "__main__":
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 30)
