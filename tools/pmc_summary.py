#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc sqlite output: mean counter value per kernel.

    python tools/pmc_summary.py <results.db> [name-filter]

Schema-introspects the rocpd_* tables (per-session suffix, like
tools/trace_summary.py); falls back to dumping the schema if the layout
is unexpected.
"""

import sqlite3
import sys
from collections import defaultdict


def main():
    db = sys.argv[1]
    filt = sys.argv[2] if len(sys.argv) > 2 else ""
    con = sqlite3.connect(db)
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    try:
        sfx = next(t.split("rocpd_kernel_dispatch_")[1] for t in tables
                   if t.startswith("rocpd_kernel_dispatch_"))
    except StopIteration:
        print("no rocpd_kernel_dispatch table; tables:", tables)
        return 1
    pmc = next((t for t in tables if t.startswith("rocpd_pmc_event_")), None)
    if pmc is None:
        print("no pmc table; tables:", tables)
        return 1
    cols = {t: [c[1] for c in con.execute(f"PRAGMA table_info({t})")]
            for t in (pmc, f"rocpd_kernel_dispatch_{sfx}",
                      f"rocpd_info_kernel_symbol_{sfx}")}
    kd = f"rocpd_kernel_dispatch_{sfx}"
    ks = f"rocpd_info_kernel_symbol_{sfx}"
    pd = next((t for t in tables if t.startswith("rocpd_info_pmc_")), None)
    try:
        q = f"""
        SELECT s.display_name, i.name, AVG(p.value), COUNT(*)
        FROM {pmc} p
        JOIN {kd} d ON p.event_id = d.event_id
        JOIN {ks} s ON d.kernel_id = s.id
        JOIN {pd} i ON p.pmc_id = i.id
        GROUP BY s.display_name, i.name"""
        rows = list(con.execute(q))
    except sqlite3.OperationalError:
        # layout differs -- dump schema so the query can be adapted
        for t, cs in cols.items():
            print(t, cs)
        if pd:
            print(pd, [c[1] for c in con.execute(f"PRAGMA table_info({pd})")])
        ex = con.execute(f"SELECT * FROM {pmc} LIMIT 3").fetchall()
        print("sample pmc rows:", ex)
        return 1
    agg = defaultdict(dict)
    for kname, cname, val, n in rows:
        if filt and filt not in kname:
            continue
        agg[kname[:60]][cname] = (val, n)
    for kname in sorted(agg):
        parts = "  ".join(f"{c}={v:.2f}(n={n})"
                          for c, (v, n) in sorted(agg[kname].items()))
        print(f"{kname:62s} {parts}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
