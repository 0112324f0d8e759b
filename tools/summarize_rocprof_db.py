#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a small text table (top kernels in
the steady-state window = last --window-ms of the timeline)."""
import argparse
import glob
import sqlite3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("db")
    p.add_argument("out")
    p.add_argument("--window-ms", type=float, default=None,
                   help="restrict to the last N ms of the timeline")
    p.add_argument("--top", type=int, default=30)
    args = p.parse_args()
    con = sqlite3.connect(args.db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith("rocpd_kernel_dispatch_")][0]
    sfx = kd[len("rocpd_kernel_dispatch_"):]
    ks = f"rocpd_info_kernel_symbol_{sfx}"
    cond = ""
    if args.window_ms:
        tmax = cur.execute(f"SELECT MAX(end) FROM {kd}").fetchone()[0]
        cond = f"WHERE k.start > {tmax - int(args.window_ms * 1e6)}"
    rows = list(cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3
        FROM {kd} k JOIN {ks} s ON k.kernel_id = s.id {cond}
        GROUP BY s.display_name ORDER BY 3 DESC LIMIT {args.top}"""))
    tot = cur.execute(
        f"SELECT SUM(k.end-k.start)/1e6 FROM {kd} k {cond}").fetchone()[0]
    span = cur.execute(
        f"SELECT (MAX(k.end)-MIN(k.start))/1e6 FROM {kd} k {cond}"
    ).fetchone()[0]
    with open(args.out, "w") as f:
        f.write(f"window {span:.0f} ms wall, {tot:.1f} ms kernel time\n")
        f.write(f"{'total_ms':>9} {'calls':>6} {'avg_us':>8}  kernel\n")
        for name, n, ms, us in rows:
            f.write(f"{ms:9.1f} {n:6} {us:8.1f}  {name[:110]}\n")


if __name__ == "__main__":
    main()
