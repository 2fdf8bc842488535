#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel-trace SQLite db into a compact per-kernel
table (run on the GPU box right after rocprofv3; the raw db is too big to
ship back).

Usage: python tools/prof_summary.py <db-glob> <out.txt> [--steps N]
Steady state is taken as everything after the last naive_conv dispatch
(MIOpen find/tuning phase)."""

import argparse
import glob
import sqlite3
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db_glob")
    ap.add_argument("out")
    ap.add_argument("--steps", type=float, default=None,
                    help="divide totals by this many steps")
    ap.add_argument("--top", type=int, default=40)
    args = ap.parse_args()

    dbs = sorted(glob.glob(args.db_glob, recursive=True))
    if not dbs:
        print(f"no db matches {args.db_glob}", file=sys.stderr)
        sys.exit(1)
    con = sqlite3.connect(dbs[-1])
    cur = con.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    kt = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0]
    sfx = kt[len("rocpd_kernel_dispatch_"):]

    cut = list(cur.execute(
        f"""SELECT MAX(kd.end) FROM rocpd_kernel_dispatch_{sfx} kd
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
            WHERE ks.display_name LIKE 'naive_conv%'"""))[0][0] or 0

    rows = list(cur.execute(
        f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
                   AVG(kd.end-kd.start)/1e3
            FROM rocpd_kernel_dispatch_{sfx} kd
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
            WHERE kd.start > {cut}
            GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {args.top}"""))
    tot = list(cur.execute(
        f"""SELECT SUM(end-start)/1e6, COUNT(*), (MAX(end)-MIN(start))/1e6
            FROM rocpd_kernel_dispatch_{sfx} WHERE start > {cut}"""))[0]

    with open(args.out, "w") as fh:
        fh.write(f"db: {dbs[-1]}\n")
        fh.write(f"steady-state kernel time {tot[0]:.1f} ms over wall {tot[2]:.1f} ms, "
                 f"{tot[1]} dispatches\n")
        if args.steps:
            fh.write(f"normalized per step (/{args.steps:g}):\n")
        fh.write(f"{'ms':>10} {'ms/step' if args.steps else '':>9} {'count':>7} "
                 f"{'avg_us':>8}  kernel\n")
        for name, cnt, ms, us in rows:
            per = f"{ms/args.steps:9.3f}" if args.steps else "         "
            fh.write(f"{ms:10.2f} {per} {cnt:7d} {us:8.1f}  {name[:110]}\n")
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
