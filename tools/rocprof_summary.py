#!/usr/bin/env python3
"""Summarize rocprofv3 sqlite output (.db) into per-kernel stats and PMC
per-dispatch values. Usage:

  python tools/rocprof_summary.py gpurun_out/prof/stats1_results.db
  python tools/rocprof_summary.py gpurun_out/prof/pmc_fetch_results.db

Used to produce profiles/rocprof_r*_summary.md. Remember the gfx950
FETCH_SIZE halving for wide coalesced reads (MI355X_MICROARCH.md §HBM):
double FETCH_SIZE before comparing with algorithmic bytes."""
import re
import sqlite3
import sys


def tables(con):
    return {re.sub(r"_0000.*", "", r[0]): r[0]
            for r in con.execute(
                "SELECT name FROM sqlite_master WHERE type IN "
                "('table','view')")}


def main(path):
    con = sqlite3.connect(path)
    t = tables(con)
    kd, ks = t["rocpd_kernel_dispatch"], t["rocpd_info_kernel_symbol"]
    print("== kernel dispatch stats ==")
    q = (f"SELECT sym.display_name, COUNT(*), AVG(k.end-k.start)/1e6, "
         f"SUM(k.end-k.start)/1e6 FROM {kd} k "
         f"JOIN {ks} sym ON k.kernel_id=sym.id GROUP BY 1 ORDER BY 4 DESC")
    for name, n, avg, tot in con.execute(q):
        print(f"  n={n:5d} avg={avg:9.3f}ms tot={tot:10.1f}ms  "
              f"{str(name)[:70]}")
    if "rocpd_pmc_event" in t:
        pe = t["rocpd_pmc_event"]
        try:
            q = (f"SELECT sym.display_name, COUNT(*), AVG(p.value), "
                 f"SUM(p.value) FROM {pe} p "
                 f"JOIN {kd} k ON p.event_id = k.event_id "
                 f"JOIN {ks} sym ON k.kernel_id = sym.id "
                 f"GROUP BY 1 ORDER BY 4 DESC")
            rows = list(con.execute(q))
            if rows:
                print("== PMC per-dispatch (counter units, usually KB) ==")
                for name, n, avg, tot in rows:
                    print(f"  n={n:4d} avg={avg:18.1f}  {str(name)[:60]}")
        except sqlite3.Error as e:
            print("pmc query failed:", e)


if __name__ == "__main__":
    for p in sys.argv[1:]:
        print(f"### {p}")
        main(p)
