#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel-trace SQLite DB into text (run ON the GPU
box so only the summary travels back through gpurun_out's 64 MiB cap).

Usage: python tools/prof_summary.py <results.db> [--delim sample] [--last N]

Picks step windows delimited by kernels whose name contains --delim
(default: the fused sampling kernel, i.e. one decode step per window) and
prints wall/busy/gap plus the per-kernel time breakdown of a
representative window, then whole-run totals.
"""
from __future__ import annotations

import argparse
import sqlite3


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--delim", default="sample")
    ap.add_argument("--last", type=int, default=3)
    ap.add_argument("--top", type=int, default=16)
    args = ap.parse_args()

    c = sqlite3.connect(args.db)
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kt = [t for t in tabs if t.startswith("rocpd_kernel_dispatch")][0]
    sfx = kt[len("rocpd_kernel_dispatch_"):]
    rows = list(c.execute(f"""
        SELECT d.start, d.end, s.display_name, d.grid_size_x
        FROM rocpd_kernel_dispatch_{sfx} d
        JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id = s.id
        ORDER BY d.start"""))
    print(f"dispatches: {len(rows)}  "
          f"span: {(rows[-1][1] - rows[0][0]) / 1e6:.1f} ms  "
          f"busy: {sum(e - s for s, e, *_ in rows) / 1e6:.1f} ms")

    marks = [i for i, r in enumerate(rows) if args.delim in r[2]]
    if len(marks) < 3:
        print(f"(fewer than 3 '{args.delim}' delimiters; totals only)")
        marks = []
    for wi in range(max(0, len(marks) - args.last - 1), len(marks) - 1):
        a, b = marks[wi], marks[wi + 1]
        seg = rows[a + 1:b + 1]
        wall = (rows[b][1] - rows[a][1]) / 1e6
        busy = sum(e - s for s, e, *_ in seg) / 1e6
        print(f"\nwindow {wi}: wall {wall:.3f} ms busy {busy:.3f} "
              f"gap {wall - busy:.3f}")
        agg = {}
        for s, e, n, gx in seg:
            k = (n.split("(")[0][:60], gx)
            t0, cnt = agg.get(k, (0, 0))
            agg[k] = (t0 + (e - s), cnt + 1)
        for (k, gx), (t0, cnt) in sorted(agg.items(),
                                         key=lambda kv: -kv[1][0])[:args.top]:
            print(f"  {t0 / 1e6:8.3f} ms {cnt:4d}x grid {gx:<9d} {k}")

    print("\nwhole-run totals:")
    agg = {}
    for s, e, n, gx in rows:
        k = n.split("(")[0][:60]
        t0, cnt = agg.get(k, (0, 0))
        agg[k] = (t0 + (e - s), cnt + 1)
    for k, (t0, cnt) in sorted(agg.items(), key=lambda kv: -kv[1][0])[:args.top]:
        print(f"  {t0 / 1e6:9.2f} ms {cnt:6d}x {k}")


if __name__ == "__main__":
    main()
