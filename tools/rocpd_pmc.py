#!/usr/bin/env python3
"""Per-kernel PMC counter table from a rocprofv3 --pmc rocpd database.

    rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY ... -d DIR -o NAME -- <cmd>
    python tools/rocpd_pmc.py DIR/*_results.db

Prints, for each kernel (sorted by SQ_WAVE_CYCLES): the counter sums and
the derived waits breakdown (wave-parked vs issue-stall vs active).
"""

import glob
import sqlite3
import sys


def summarize(path):
    con = sqlite3.connect(path)
    c = con.cursor()
    tabs = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    ev = next(t for t in tabs if t.startswith("rocpd_pmc_event"))
    info = next(t for t in tabs if t.startswith("rocpd_info_pmc"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    rows = c.execute(f"""
        SELECT s.display_name, i.name, SUM(e.value)
        FROM {ev} e JOIN {disp} d ON e.event_id = d.event_id
        JOIN {sym} s ON d.kernel_id = s.id
        JOIN {info} i ON e.pmc_id = i.id
        GROUP BY s.display_name, i.name""").fetchall()
    con.close()
    agg = {}
    for name, cname, val in rows:
        agg.setdefault(name.replace("(anonymous namespace)::", "")[:80], {})[cname] = val
    print(f"# {path}")
    key = "SQ_WAVE_CYCLES"
    for name, cs in sorted(agg.items(), key=lambda kv: -kv[1].get(key, 0))[:16]:
        print(name)
        wc = cs.get(key, 0)
        for cname, val in sorted(cs.items()):
            pct = f"  ({100*val/wc:5.1f}% of wave cycles)" \
                if wc and cname.startswith("SQ_WAIT") or cname == "SQ_ACTIVE_INST_ANY" \
                and wc else ""
            print(f"   {cname:24s} {val:16.3e}{pct}")


if __name__ == "__main__":
    for arg in sys.argv[1:]:
        for p in glob.glob(arg):
            summarize(p)
