#!/usr/bin/env python3
"""Summarize a rocprofv3 results db (kernel-trace) into a markdown table.

Usage: python tools/profsummary.py gpurun_out/profN/*.db [steps] > profiles/...md
"""

import sqlite3
import sys


def summarize(db_path, steps=None):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")]
    if not t:
        print("no kernel dispatch table found", file=sys.stderr)
        return
    sfx = t[0].replace("rocpd_kernel_dispatch_", "")
    rows = list(cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
               AVG(k.end-k.start)/1e3,
               MAX(ks.arch_vgpr_count), MAX(ks.accum_vgpr_count),
               MAX(ks.group_segment_size)
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC"""))
    total = sum(r[2] for r in rows)
    print(f"# Kernel profile: {db_path}")
    print()
    if steps:
        print(f"{steps} training steps profiled; total kernel time "
              f"{total:.2f} ms = {total/steps*1000:.0f} us/step")
        print()
    print("| total ms | calls | us/call | VGPR | AGPR | LDS B | kernel |")
    print("|---|---|---|---|---|---|---|")
    for name, cnt, ms, us, vgpr, agpr, lds in rows:
        print(f"| {ms:.3f} | {cnt} | {us:.2f} | {vgpr} | {agpr} | {lds} | "
              f"`{str(name)[:80]}` |")


if __name__ == "__main__":
    summarize(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else None)
