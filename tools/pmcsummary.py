#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results db: per-kernel wave-state buckets.

Usage: python tools/pmcsummary.py gpurun_out/.../x_results.db
(counters expected among: SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY
SQ_WAIT_INST_LDS SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES)
"""

import collections
import sqlite3
import sys


def main():
    db = sys.argv[1]
    con = sqlite3.connect(db)
    cur = con.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")][0]
    sfx = t.replace("rocpd_kernel_dispatch_", "")
    q = (f"SELECT ks.display_name, pi.name, SUM(pe.value) "
         f"FROM rocpd_pmc_event_{sfx} pe "
         f"JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id=pi.id "
         f"JOIN rocpd_kernel_dispatch_{sfx} k ON pe.event_id=k.id "
         f"JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id "
         f"GROUP BY 1,2")
    d = collections.defaultdict(dict)
    for kn, cn, v in cur.execute(q):
        d[kn.split("(")[0][:44]][cn] = v
    print(f"{'kernel':44s}  wave(G)  wait%  instw%  lds%  act%  mfma%")
    for k, row in sorted(d.items(),
                         key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", 0)):
        wc = row.get("SQ_WAVE_CYCLES", 0)
        if wc < 1e7:
            continue
        def pct(name, scale=1):
            return 100.0 * row.get(name, 0) / (scale * wc)
        print(f"{k:44s}  {wc/1e9:6.2f}  {pct('SQ_WAIT_ANY'):5.1f}  "
              f"{pct('SQ_WAIT_INST_ANY'):5.1f}  "
              f"{pct('SQ_WAIT_INST_LDS'):5.1f}  "
              f"{pct('SQ_ACTIVE_INST_ANY'):5.1f}  "
              f"{pct('SQ_VALU_MFMA_BUSY_CYCLES', 4):5.1f}")


if __name__ == "__main__":
    main()
