#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd results.db into a per-kernel stats table.

    python tools/rocpd_stats.py gpurun_out/prof/runc/*_results.db [-n 25]

Prints kernels sorted by total GPU time: % of total, calls, avg us, name.
(rocprofv3 on this image emits the SQL database only; this replaces the
old text --stats summary so profile evidence can be committed under
profiles/.)
"""

import argparse
import glob
import sqlite3


def summarize(path, top):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(
        f"SELECT s.display_name, COUNT(*), SUM(d.end - d.start) "
        f"FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id "
        f"GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"
    ).fetchall()
    total = sum(r[2] for r in rows) or 1
    print(f"# {path}")
    print(f"total GPU time: {total / 1e6:.2f} ms over "
          f"{sum(r[1] for r in rows)} dispatches")
    print(f"{'%':>5s} {'calls':>6s} {'avg_us':>8s}  kernel")
    for name, calls, ns in rows[:top]:
        print(f"{100.0 * ns / total:5.1f} {calls:6d} {ns / calls / 1e3:8.1f}"
              f"  {name[:100]}")
    db.close()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("dbs", nargs="+")
    ap.add_argument("-n", type=int, default=25)
    args = ap.parse_args()
    for pat in args.dbs:
        for path in sorted(glob.glob(pat)) or [pat]:
            summarize(path, args.n)
            print()


if __name__ == "__main__":
    main()
