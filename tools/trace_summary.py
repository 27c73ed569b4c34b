#!/usr/bin/env python3
"""Summarize a rocprofv3 --kernel-trace results.db into a per-kernel table.

rocprofv3 writes rocpd_* tables with a per-session suffix; this resolves the
suffix, optionally restricts to the steady-state tail of the timeline (the
first part of a bench run is warmup + hipBLASLt autotune sweep), and prints
a markdown table plus a per-window roll-up.

    python tools/trace_summary.py DB [--tail 0.5] [--windows N]
"""

import argparse
import sqlite3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("db")
    p.add_argument("--tail", type=float, default=0.5,
                   help="analyze only the last FRACTION of the kernel timeline")
    p.add_argument("--windows", type=int, default=0,
                   help="divide totals by N (e.g. timed accumulation windows)")
    p.add_argument("--limit", type=int, default=30)
    args = p.parse_args()

    db = sqlite3.connect(args.db)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = next(t.split("rocpd_kernel_dispatch_")[1] for t in tables
               if t.startswith("rocpd_kernel_dispatch_"))
    kd, ks = f"rocpd_kernel_dispatch_{sfx}", f"rocpd_info_kernel_symbol_{sfx}"

    t0, t1 = cur.execute(f"SELECT MIN(start), MAX(end) FROM {kd}").fetchone()
    cut = t1 - (t1 - t0) * args.tail
    q = f"""
    SELECT k.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e3
    FROM {kd} d JOIN {ks} k ON d.kernel_id = k.id
    WHERE d.start >= ? GROUP BY k.display_name
    ORDER BY SUM(d.end-d.start) DESC
    """
    rows = cur.execute(q, (cut,)).fetchall()
    tot = sum(r[2] for r in rows)
    span_ms = (t1 - cut) / 1e6
    print(f"analyzed tail: {span_ms:.1f} ms wall, {tot:.1f} ms kernel time, "
          f"{sum(r[1] for r in rows)} dispatches")
    print("| total ms | % | n | avg us |" +
          (" us/window |" if args.windows else "") + " kernel |")
    print("|---:|---:|---:|---:|" + ("---:|" if args.windows else "") + ":---|")
    shown = 0.0
    for name, n, ms, avg in rows[: args.limit]:
        w = f" {ms * 1000 / args.windows:.1f} |" if args.windows else ""
        short = name if len(name) < 72 else name[:69] + "..."
        print(f"| {ms:.2f} | {100 * ms / tot:.1f} | {n} | {avg:.2f} |{w} `{short}` |")
        shown += ms
    rest = tot - shown
    if rest > 0.005:
        print(f"| {rest:.2f} | {100 * rest / tot:.1f} | | |" +
              (" |" if args.windows else "") + " (all others) |")

    # pool roll-up: GEMMs vs custom kernels
    pools = {}
    for name, n, ms, avg in rows:
        key = ("hipBLASLt GEMM" if name.startswith("Cijk_") else
               name if name.startswith("k_") else "other")
        a = pools.setdefault(key, [0, 0.0])
        a[0] += n
        a[1] += ms
    print("\npools:")
    for key, (n, ms) in sorted(pools.items(), key=lambda kv: -kv[1][1]):
        w = f"  ({ms * 1000 / args.windows:.1f} us/window)" if args.windows else ""
        print(f"  {ms:8.2f} ms  n={n:6d}  {key}{w}")


if __name__ == "__main__":
    main()
