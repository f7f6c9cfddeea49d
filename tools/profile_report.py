#!/usr/bin/env python3
"""Turn a rocprofv3 results DB into the per-kernel markdown table used in
profiles/ (see profiles/r01_bench_profile.md).

Usage: python tools/profile_report.py gpurun_out/profX/whatever_results.db \
           [--steps N] [--top K]
"""

import argparse
import sqlite3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--steps", type=int, default=15,
                    help="bench steps in the trace (for per-step totals)")
    ap.add_argument("--top", type=int, default=20)
    ap.add_argument("--by-grid", action="store_true",
                    help="also break the conv kernels down by grid shape")
    args = ap.parse_args()

    db = sqlite3.connect(args.db)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    if not tables:
        raise SystemExit("no kernel dispatch table in this DB")
    sfx = tables[0][len("rocpd_kernel_dispatch_"):]

    q = f"""SELECT ks.display_name, COUNT(*), SUM(k.end-k.start),
                   AVG(k.end-k.start)
            FROM rocpd_kernel_dispatch_{sfx} k
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
            GROUP BY 1 ORDER BY 3 DESC LIMIT {args.top}"""
    rows = list(cur.execute(q))
    tot = sum(r[2] for r in rows)
    print("| kernel | calls/step | avg µs | % of GPU-busy |")
    print("|---|---|---|---|")
    for name, calls, tns, ans in rows:
        short = name.split("(")[0]
        print(f"| {short} | {calls / args.steps:.1f} | {ans / 1e3:.1f} "
              f"| {tns / tot * 100:.1f}% |")
    print(f"\nGPU-busy per step: {tot / 1e6 / args.steps:.2f} ms "
          f"(streams serialized in the trace; wall step is lower when the "
          f"teacher/dW side streams overlap)")

    if args.by_grid:
        for kname in ("conv2d_fwd", "conv2d_bwd_weight", "conv2d_bwd_data"):
            q2 = f"""SELECT k.grid_size_x/256, k.grid_size_y, k.grid_size_z,
                            COUNT(*), SUM(k.end-k.start)/1e6,
                            AVG(k.end-k.start)/1e3
                     FROM rocpd_kernel_dispatch_{sfx} k
                     JOIN rocpd_info_kernel_symbol_{sfx} ks
                          ON k.kernel_id = ks.id
                     WHERE ks.display_name LIKE '%{kname}%'
                     GROUP BY 1,2,3 ORDER BY 5 DESC LIMIT 6"""
            print(f"\n### {kname} by grid")
            for gx, gy, gz, calls, tt, avg in cur.execute(q2):
                print(f"  grid=({gx},{gy},{gz}) calls={calls} "
                      f"total={tt:.2f}ms avg={avg:.1f}µs")


if __name__ == "__main__":
    main()
