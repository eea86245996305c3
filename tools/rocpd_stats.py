#!/usr/bin/env python3
"""Summarize a rocprofv3 SQLite results DB: top kernels by total duration.

Usage: python tools/rocpd_stats.py results.db [-n 30] [-o out.csv]
Schema-flexible: finds the kernel-dispatch table by inspecting sqlite_master.
"""
import argparse
import csv
import sqlite3
import sys


def find_kernel_view(cur):
    cur.execute("SELECT name, type FROM sqlite_master "
                "WHERE type IN ('table','view')")
    names = [r[0] for r in cur.fetchall()]
    # preferred: rocprofv3 rocpd schema has a kernel dispatch view/table
    for cand in names:
        try:
            cur.execute(f"PRAGMA table_info({cand})")
            cols = [r[1].lower() for r in cur.fetchall()]
        except sqlite3.Error:
            continue
        has_name = any(c in cols for c in
                       ("kernel_name", "name", "kernelname"))
        has_time = ("start" in cols and "end" in cols) or \
            any("duration" in c for c in cols)
        if ("kernel" in cand.lower() or "dispatch" in cand.lower()) and \
                has_name and has_time:
            yield cand, cols


def summarize(db_path, topn=30, tail_frac=0.0):
    """tail_frac > 0: only count dispatches whose start falls in the LAST
    `tail_frac` of the trace window (drops MIOpen-find / warmup noise)."""
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    rows = None
    for table, cols in find_kernel_view(cur):
        name_col = next(c for c in ("kernel_name", "name", "kernelname")
                        if c in cols)
        if not ("start" in cols and "end" in cols):
            continue
        where = ""
        if tail_frac > 0:
            cur.execute(f"SELECT MIN(start), MAX(end) FROM {table}")
            t0, t1 = cur.fetchone()
            cut = t1 - (t1 - t0) * tail_frac
            where = f" WHERE start >= {cut}"
        try:
            cur.execute(
                f"SELECT {name_col}, SUM(end - start) total, COUNT(*) calls, "
                f"AVG(end - start) avg FROM {table}{where} "
                f"GROUP BY {name_col} ORDER BY total DESC LIMIT {topn}")
            rows = cur.fetchall()
            if rows:
                print(f"# table: {table} tail_frac={tail_frac}",
                      file=sys.stderr)
                break
        except sqlite3.Error:
            continue
    con.close()
    return rows or []


def main():
    p = argparse.ArgumentParser()
    p.add_argument("db")
    p.add_argument("-n", type=int, default=30)
    p.add_argument("-o", default=None)
    p.add_argument("--tail-frac", type=float, default=0.0)
    args = p.parse_args()
    rows = summarize(args.db, args.n, args.tail_frac)
    total = sum(r[1] for r in rows) or 1
    out = [("kernel", "total_ns", "calls", "avg_ns", "pct_of_top")]
    for name, tot, calls, avg in rows:
        out.append((name[:120], int(tot), int(calls), int(avg),
                    round(100 * tot / total, 2)))
    if args.o:
        with open(args.o, "w", newline="") as f:
            csv.writer(f).writerows(out)
        print(f"wrote {args.o} ({len(out) - 1} kernels)")
    else:
        for r in out:
            print(",".join(str(x) for x in r))


if __name__ == "__main__":
    main()
