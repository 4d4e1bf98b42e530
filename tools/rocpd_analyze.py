#!/usr/bin/env python3
"""Summarize rocprofv3 rocpd SQLite outputs: per-kernel dispatch stats and
(for --pmc runs) per-kernel counter sums/averages.

Usage: python tools/rocpd_analyze.py gpurun_out/prof/*.db
"""
import sqlite3
import sys


def suffix(con, base):
    for (n,) in con.execute(
            "SELECT name FROM sqlite_master WHERE type='table'"):
        if n.startswith(base):
            return n[len(base):]
    raise SystemExit(f"no {base} table")


def analyze(db):
    con = sqlite3.connect(db)
    s = suffix(con, "rocpd_kernel_dispatch")
    print(f"\n===== {db}")
    q = f"""
      SELECT k.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
             AVG(d.end-d.start)/1e3,
             k.arch_vgpr_count, k.sgpr_count, d.group_segment_size,
             d.grid_size_x, d.workgroup_size_x
      FROM rocpd_kernel_dispatch{s} d
      JOIN rocpd_info_kernel_symbol{s} k ON k.id = d.kernel_id
      GROUP BY k.display_name ORDER BY SUM(d.end-d.start) DESC"""
    print(f"{'kernel':<34}{'count':>6}{'tot_ms':>10}{'avg_us':>10}"
          f"{'vgpr':>6}{'sgpr':>6}{'lds':>8}{'grid':>10}{'wg':>5}")
    for row in con.execute(q):
        name = row[0].split("(")[0][:33]
        print(f"{name:<34}{row[1]:>6}{row[2]:>10.3f}{row[3]:>10.1f}"
              f"{row[4]:>6}{row[5]:>6}{row[6]:>8}{row[7]:>10}{row[8]:>5}")
    # PMC values, if any
    try:
        pmc_info = dict(con.execute(
            f"SELECT id, name FROM rocpd_info_pmc{s}"))
    except sqlite3.OperationalError:
        pmc_info = {}
    if pmc_info:
        q = f"""
          SELECT k.display_name, p.pmc_id, SUM(p.value), AVG(p.value),
                 COUNT(*)
          FROM rocpd_pmc_event{s} p
          JOIN rocpd_kernel_dispatch{s} d ON d.event_id = p.event_id
          JOIN rocpd_info_kernel_symbol{s} k ON k.id = d.kernel_id
          GROUP BY k.display_name, p.pmc_id"""
        try:
            rows = list(con.execute(q))
        except sqlite3.OperationalError as e:
            print("pmc join failed:", e)
            rows = []
        for name, pmc_id, tot, avg, cnt in rows:
            print(f"  PMC {pmc_info.get(pmc_id, pmc_id):<26} "
                  f"{name.split('(')[0][:30]:<32} sum={tot:.4g} "
                  f"avg/disp={avg:.4g} n={cnt}")


if __name__ == "__main__":
    for db in sys.argv[1:]:
        analyze(db)
