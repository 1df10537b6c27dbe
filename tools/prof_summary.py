#!/usr/bin/env python3
"""Summarize a rocprofv3 results db (kernel-trace) into a text table for
profiles/ (committed evidence). Usage:
    python tools/prof_summary.py gpurun_out/prof2 > profiles/rNN_topk.txt
"""

import glob
import sqlite3
import sys


def main(path):
    dbs = glob.glob(f"{path}/**/*.db", recursive=True) or glob.glob(path)
    assert dbs, f"no results db under {path}"
    for db in dbs:
        c = sqlite3.connect(db)
        tables = [r[0] for r in c.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        kd = [t for t in tables if t.startswith("rocpd_kernel_dispatch")]
        if not kd:
            continue
        sfx = kd[0][len("rocpd_kernel_dispatch_"):]
        print(f"# {db}")
        print(f"{'kernel':62s} {'n':>5s} {'total_ms':>10s} {'avg_ms':>9s} "
              f"{'min_ms':>8s} {'max_ms':>8s} {'grid':>9s} {'lds':>7s} "
              f"{'vgpr':>5s}")
        q = f"""
        SELECT ks.kernel_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e6, MIN(k.end-k.start)/1e6,
               MAX(k.end-k.start)/1e6, MAX(k.grid_size_x),
               MAX(k.group_segment_size), MAX(ks.arch_vgpr_count)
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.kernel_name ORDER BY 3 DESC
        """
        for r in c.execute(q):
            name = r[0].replace(".kd", "")[:62]
            print(f"{name:62s} {r[1]:5d} {r[2]:10.3f} {r[3]:9.3f} "
                  f"{r[4]:8.3f} {r[5]:8.3f} {r[6]:9d} {r[7] or 0:7d} "
                  f"{r[8] or 0:5d}")
        print()


if __name__ == "__main__":
    main(sys.argv[1])
