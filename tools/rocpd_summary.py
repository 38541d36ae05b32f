#!/usr/bin/env python3
"""Summarize rocprofv3 rocpd SQLite outputs (kernel stats / PMC counters).

Usage:
  python tools/rocpd_summary.py trace  <results.db>   # per-kernel durations
  python tools/rocpd_summary.py pmc    <results.db>   # per-kernel counter sums
"""
import sqlite3
import sys


def tables(cur):
    return [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]


def find(ts, frag):
    return [t for t in ts if frag in t][0]


def main():
    mode, db = sys.argv[1], sys.argv[2]
    con = sqlite3.connect(db)
    cur = con.cursor()
    ts = tables(cur)
    disp, sym = find(ts, "kernel_dispatch"), find(ts, "kernel_symbol")
    if mode == "trace":
        q = f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e9,
                 AVG(d.end-d.start)/1e6, MIN(d.end-d.start)/1e6, MAX(d.end-d.start)/1e6
                 FROM {disp} d JOIN {sym} s ON s.id = d.kernel_id
                 GROUP BY s.display_name ORDER BY 3 DESC"""
        print("%-52s %6s %10s %10s %10s %10s" % ("kernel", "count", "total_s", "avg_ms", "min_ms", "max_ms"))
        for r in cur.execute(q):
            print("%-52s %6d %10.3f %10.3f %10.3f %10.3f" % (r[0][:52], *r[1:]))
    elif mode == "pmc":
        pmc, info = find(ts, "pmc_event"), find(ts, "info_pmc")
        q = f"""SELECT s.display_name, i.name, COUNT(*), SUM(p.value), AVG(p.value)
                 FROM {pmc} p JOIN {disp} d ON d.event_id = p.event_id
                 JOIN {sym} s ON s.id = d.kernel_id
                 JOIN {info} i ON i.id = p.pmc_id
                 GROUP BY s.display_name, i.name ORDER BY 1, 2"""
        for r in cur.execute(q):
            print("%-48s %-22s n=%4d sum=%16.0f avg=%14.1f" % (r[0][:48], r[1], r[2], r[3] or 0, r[4] or 0))
    else:
        raise SystemExit("unknown mode")


if __name__ == "__main__":
    main()
