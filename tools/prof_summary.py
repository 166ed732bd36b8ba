#!/usr/bin/env python3
"""Summarize a rocprofv3 sqlite results DB: per-kernel totals + timeline
busy/gap analysis (launch-bound vs kernel-bound).  Run on the GPU box right
after profiling; commit the text output under profiles/."""
import re
import sqlite3
import sys


def main(path, steady_frac=0.5):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))

    rows = cur.execute(f"""
        SELECT ks.kernel_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3
        FROM {kd} k JOIN {ks} ks ON k.kernel_id = ks.id
        GROUP BY ks.kernel_name ORDER BY 3 DESC LIMIT 40""").fetchall()
    total, span = cur.execute(
        f"SELECT SUM(end-start)/1e6, (MAX(end)-MIN(start))/1e6 FROM {kd}"
    ).fetchone()
    print(f"{'total ms':>9} {'count':>7} {'avg us':>9}  kernel")
    for name, n, ms, us in rows:
        print(f"{ms:9.2f} {n:7d} {us:9.1f}  {re.sub(r'[(].*', '', name)[:95]}")
    print(f"\nALL kernels: sum={total:.1f} ms, wall span={span:.1f} ms")

    # steady-state gap analysis on the tail of the timeline
    iv = cur.execute(f"SELECT start, end FROM {kd} ORDER BY start").fetchall()
    t0 = iv[0][0] + (iv[-1][1] - iv[0][0]) * steady_frac
    tail = [(s, e) for s, e in iv if s >= t0]
    busy = 0
    cs, ce = tail[0]
    for s, e in tail[1:]:
        if s > ce:
            busy += ce - cs
            cs, ce = s, e
        else:
            ce = max(ce, e)
    busy += ce - cs
    wall = tail[-1][1] - tail[0][0]
    print(f"steady tail ({steady_frac:.0%}..): wall {wall / 1e6:.1f} ms, "
          f"kernel-busy {busy / 1e6:.1f} ms "
          f"({busy / wall * 100:.1f}%), gaps {(wall - busy) / 1e6:.1f} ms, "
          f"dispatches {len(tail)}")

    # densest 200 ms window over the WHOLE timeline: the tail window above
    # can straddle idle phases outside the timed region (warmup sync, algo
    # lookups, teardown), under-reporting busy%; this is the honest
    # steady-state number for a bench whose timed region is bursts of steps
    win = 200_000_000  # ns
    merged = []
    cs, ce = iv[0]
    for s, e in iv[1:]:
        if s > ce:
            merged.append((cs, ce))
            cs, ce = s, e
        else:
            ce = max(ce, e)
    merged.append((cs, ce))
    best = 0.0
    j = 0
    pref = [0]
    for s, e in merged:
        pref.append(pref[-1] + (e - s))
    for i, (s, _e) in enumerate(merged):
        hi = s + win
        j = i
        acc = 0
        while j < len(merged) and merged[j][0] < hi:
            acc += min(merged[j][1], hi) - merged[j][0]
            j += 1
        best = max(best, acc / win)
    print(f"densest 200 ms window: kernel-busy {best * 100:.1f}%")


if __name__ == "__main__":
    main(sys.argv[1], float(sys.argv[2]) if len(sys.argv) > 2 else 0.5)
