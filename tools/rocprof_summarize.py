#!/usr/bin/env python3
"""Summarize rocprofv3 result databases (rocpd sqlite) into the committed
profile artifacts under profiles/.

Usage: python tools/rocprof_summarize.py <results.db> [...] > profiles/xxx.md
"""

import glob
import sqlite3
import sys


def suffix_for(con, prefix):
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    cands = [t for t in tabs if t.startswith(prefix)]
    return cands[0].replace(prefix + "_", "") if cands else None


def summarize(path):
    con = sqlite3.connect(path)
    print("## %s" % path)
    sfx = suffix_for(con, "rocpd_kernel_dispatch")
    if sfx:
        q = f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
               AVG(kd.end-kd.start)/1e6,
               MAX(ks.arch_vgpr_count), MAX(ks.accum_vgpr_count),
               MAX(ks.sgpr_count), MAX(ks.group_segment_size)
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 12"""
        rows = list(con.execute(q))
        if rows:
            print("\n| kernel | n | total ms | avg ms | vgpr | agpr | sgpr | lds B |")
            print("|---|---|---|---|---|---|---|---|")
            for name, n, tot, avg, v, a, s, l in rows:
                short = str(name).replace("(anonymous namespace)::", "")
                short = short.split("(")[0][-60:] or short[:60]
                print("| %s | %d | %.3f | %.4f | %s | %s | %s | %s |"
                      % (short, n, tot, avg, v, a, s, l))
    sfx = suffix_for(con, "rocpd_pmc_event")
    if sfx:
        q = f"""SELECT pi.name, SUM(pe.value)
        FROM rocpd_pmc_event_{sfx} pe
        JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id GROUP BY pi.name"""
        rows = list(con.execute(q))
        if rows:
            print("\n| counter | sum |")
            print("|---|---|")
            for name, val in rows:
                print("| %s | %.0f |" % (name, val))
    print()


if __name__ == "__main__":
    paths = []
    for arg in sys.argv[1:]:
        paths.extend(glob.glob(arg))
    for p in paths:
        summarize(p)
