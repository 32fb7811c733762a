#!/usr/bin/env python3
"""Measure GPU idle gaps between kernel dispatches in a rocprofv3
kernel-trace db: merges intervals across all queues/streams, reports
busy vs idle over the traced window and the largest repeating gaps.

Usage: python tools/gapscan.py gpurun_out/.../*_results.db [n_steps]
"""

import sqlite3
import sys


def main():
    db = sys.argv[1]
    nsteps = int(sys.argv[2]) if len(sys.argv) > 2 else None
    con = sqlite3.connect(db)
    cur = con.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'")]
    sfx = t[0].replace("rocpd_kernel_dispatch_", "")
    rows = list(cur.execute(f"""
        SELECT k.start, k.end, ks.display_name
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        ORDER BY k.start"""))
    if not rows:
        print("no dispatches")
        return
    # steady state = the tightest contiguous run of steps, located by the
    # one-per-step SGD apply (startup/capture/sampling phases have huge
    # host-side gaps that would swamp the replay-gap signal)
    sgd = [s for (s, e, n) in rows if n.startswith("sgd_step_kernel")]
    if len(sgd) > 25:
        run = 20
        best = min(range(len(sgd) - run),
                   key=lambda i: sgd[i + run] - sgd[i])
        lo, hi = sgd[best], sgd[best + run]
        nsteps = run
        print(f"steady window: {run} steps, "
              f"{(hi - lo) / 1e3 / run:.1f} us/step wall")
    else:
        t0, t1 = rows[0][0], rows[-1][1]
        lo = t0 + (t1 - t0) // 4
        hi = t1 - (t1 - t0) // 4
    rows = [r for r in rows if r[0] >= lo and r[1] <= hi]
    # merge busy intervals
    busy = 0
    gaps = {}   # (prev_kernel_short, next_kernel_short) -> [count, total_ns]
    cur_s, cur_e, cur_name = rows[0][0], rows[0][1], rows[0][2]
    for s, e, name in rows[1:]:
        if s <= cur_e:
            if e > cur_e:
                cur_e, cur_name = e, name
        else:
            busy += cur_e - cur_s
            key = (cur_name.split("(")[0][:28], name.split("(")[0][:28])
            g = gaps.setdefault(key, [0, 0])
            g[0] += 1
            g[1] += s - cur_e
            cur_s, cur_e, cur_name = s, e, name
    busy += cur_e - cur_s
    wall = rows[-1][1] - rows[0][0]
    idle = wall - busy
    print(f"window {wall/1e6:.3f} ms  busy {busy/1e6:.3f} ms  "
          f"idle {idle/1e6:.3f} ms ({100*idle/wall:.1f}%)")
    if nsteps:
        print(f"~{idle/1e3/nsteps:.1f} us idle per step (assuming {nsteps} "
              f"steps in the middle-half window)")
    print("\ntop gap sites (prev -> next): count, total us, us/occurrence")
    for (a, b), (n, tot) in sorted(gaps.items(), key=lambda kv: -kv[1][1])[:14]:
        print(f"  {a:28s} -> {b:28s}  {n:5d}  {tot/1e3:9.1f}  {tot/1e3/n:7.2f}")


if __name__ == "__main__":
    main()
