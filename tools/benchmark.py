#!/usr/bin/env python3
"""Local sweep driver + log scrapers — re-expression of
/root/reference/tools/benchmark.py + tf_ec2.py for one 8xMI355X node.

The reference launched EC2 fleets over SSH and scraped downloaded logs; here
the "cluster" is N ranks under torch.distributed.run on localhost, and logs
are scraped with the SAME regex contracts (step lines benchmark.py:31,
ELAPSED/ITERATION TIMES :66-68,140-142, eval lines :151).

Usage:
  python tools/benchmark.py run cfg/local_8gpu/aggregate_sweep.json
  python tools/benchmark.py use_dir cfg/local_8gpu          # all cfgs
  python tools/benchmark.py scrape /path/to/out_master      # stats only
"""

from __future__ import annotations

import json
import os
import re
import subprocess
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


# ---------------------------------------------------------------------------
# log scrapers (regex contracts shared with the reference harness)
# ---------------------------------------------------------------------------

def extract_steps(fname):
    """Max step seen (reference check_if_reached_iters, benchmark.py:24-34)."""
    cur = 0
    with open(fname) as f:
        for line in f:
            m = re.match(r".*step ([0-9]*),.*", line)
            if m:
                cur = max(cur, int(m.group(1)))
    return cur


def extract_step_records(fname):
    """[(step, loss, train_acc, examples_per_sec, sec_per_batch), ...]"""
    out = []
    pat = re.compile(r".*step ([0-9]+), loss = ([-.0-9e]+), train_acc = "
                     r"([-.0-9e]+), test_acc = ([-.0-9e]+)\(([-.0-9e]+) "
                     r"examples/sec; ([-.0-9e]+)\s+sec/batch\)")
    with open(fname) as f:
        for line in f:
            m = pat.match(line)
            if m:
                out.append((int(m.group(1)), float(m.group(2)),
                            float(m.group(3)), float(m.group(5)),
                            float(m.group(6))))
    return out


def extract_compute_times(fname):
    """Last 'ELAPSED TIMES [...]' -> [(time, worker, iteration), ...]."""
    compute_times = []
    with open(fname) as f:
        for line in f:
            m = re.match(r".*ELAPSED TIMES (.*)", line)
            if m:
                compute_times = eval(m.group(1))  # list literal, ours
    return compute_times


def extract_iteration_times(fname):
    times = []
    with open(fname) as f:
        for line in f:
            m = re.match(r".*ITERATION TIMES (.*)", line)
            if m:
                times = json.loads(m.group(1))
    return times


def extract_times_losses_precision(fname):
    """Evaluator lines (nn_eval.py:102 format)."""
    times, losses, precisions, steps = [], [], [], []
    with open(fname) as f:
        for line in f:
            m = re.match(r"Num examples: ([0-9]*)  Precision @ 1: ([.0-9]*) "
                         r"Loss: ([.0-9]*) Time: ([.0-9]*)", line)
            sm = re.match(r".* step=([0-9]*)", line)
            if m:
                times.append(float(m.group(4)))
                losses.append(float(m.group(3)))
                precisions.append(float(m.group(2)))
            if sm:
                steps.append(int(sm.group(1)))
    n = min(len(times), len(steps)) if steps else len(times)
    return times[:n], losses[:n], precisions[:n], steps[:n]


def percentile_report(compute_times):
    """benchmark.py:83-111 percentile stats over per-worker compute times."""
    if not compute_times:
        return {}
    all_times = np.array([t for (t, _, _) in compute_times])
    by_iter = {}
    for t, w, i in compute_times:
        by_iter.setdefault(i, []).append(t)
    p99 = [np.percentile(v, 99) for v in by_iter.values()]
    p95 = [np.percentile(v, 95) for v in by_iter.values()]
    p100 = [np.max(v) for v in by_iter.values()]
    return {
        "std": float(all_times.std()),
        "max": float(all_times.max()),
        "p80": float(np.percentile(all_times, 80)),
        "p90": float(np.percentile(all_times, 90)),
        "p95": float(np.percentile(all_times, 95)),
        "p99": float(np.percentile(all_times, 99)),
        "mean": float(all_times.mean()),
        "mean_p95_across_iters": float(np.mean(p95)) if p95 else None,
        "mean_p99_across_iters": float(np.mean(p99)) if p99 else None,
        "mean_p100_across_iters": float(np.mean(p100)) if p100 else None,
    }


# ---------------------------------------------------------------------------
# run driver
# ---------------------------------------------------------------------------

def run_cfg(cfg_path, outdir=None):
    with open(cfg_path) as f:
        cfg = json.load(f)
    name = cfg.get("name") or os.path.splitext(os.path.basename(cfg_path))[0]
    outdir = outdir or os.path.join(ROOT, "bench_out", name)
    os.makedirs(outdir, exist_ok=True)
    nproc = int(cfg.get("workers", 1))
    train_dir = os.path.join(outdir, "train_dir")
    flags = dict(cfg.get("flags", {}))
    flags.setdefault("train_dir", train_dir)
    flags.setdefault("synthetic_data", True)
    argv = []
    for k, v in flags.items():
        if isinstance(v, bool):
            if v:
                argv.append(f"--{k}")
        else:
            argv += [f"--{k}", str(v)]
    entry = os.path.join(ROOT, "src", "mnist_distributed_train.py")
    if nproc > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--standalone", "--local-addr", "127.0.0.1",
               f"--nproc-per-node={nproc}", entry] + argv
    else:
        cmd = [sys.executable, entry] + argv
    master_log = os.path.join(outdir, "out_master")
    # concurrent evaluator (reference run_tf launches one per sweep point,
    # tf_ec2.py:445, and benchmark.py:54-58 downloads/scrapes out_evaluator)
    eval_proc = None
    eval_log_f = None
    eval_log = os.path.join(outdir, "out_evaluator")
    if cfg.get("evaluator"):
        eargv = [sys.executable, os.path.join(ROOT, "src", "mnist_eval.py"),
                 "--checkpoint_dir", flags["train_dir"],
                 "--eval_dir", os.path.join(outdir, "eval_dir"),
                 "--eval_interval_secs",
                 str(cfg.get("eval_interval_secs", 2))]
        for k in ("synthetic_data", "fake_data"):
            if flags.get(k):
                eargv.append(f"--{k}")
        if "data_dir" in flags:
            eargv += ["--data_dir", str(flags["data_dir"])]
        if "model" in flags:
            eargv += ["--model", str(flags["model"])]
        eval_log_f = open(eval_log, "w")
        eval_proc = subprocess.Popen(eargv, stdout=eval_log_f,
                                     stderr=subprocess.STDOUT, cwd=ROOT)
    t0 = time.time()
    with open(master_log, "w") as lf:
        proc = subprocess.run(cmd, stdout=lf, stderr=subprocess.STDOUT,
                              timeout=cfg.get("timeout_s", 3600), cwd=ROOT)
    run_time = time.time() - t0
    if eval_proc is not None:
        # let the poller catch the final checkpoint, then stop THIS exact
        # child (never pattern-kills)
        deadline = time.time() + 2 * cfg.get("eval_interval_secs", 2) + 8
        while time.time() < deadline and eval_proc.poll() is None:
            time.sleep(0.5)
        if eval_proc.poll() is None:
            eval_proc.terminate()
            try:
                eval_proc.wait(15)
            except subprocess.TimeoutExpired:
                eval_proc.kill()
        eval_log_f.close()
    results = {
        "name": name,
        "cfg": cfg,
        "returncode": proc.returncode,
        "run_time": run_time,
        "max_step": extract_steps(master_log),
    }
    recs = extract_step_records(master_log)
    if recs:
        eps = np.array([r[3] for r in recs][len(recs) // 4:])
        spb = np.array([r[4] for r in recs][len(recs) // 4:])
        results["examples_per_sec_per_worker_p50"] = float(np.median(eps))
        results["sec_per_batch_p50"] = float(np.median(spb))
        results["sec_per_batch_p95"] = float(np.percentile(spb, 95))
        results["sec_per_batch_p99"] = float(np.percentile(spb, 99))
        results["final_loss"] = recs[-1][1]
        results["final_train_acc"] = recs[-1][2]
        results["node_examples_per_sec_p50"] = float(np.median(eps)) * nproc
    ct = extract_compute_times(master_log)
    if ct:
        results["compute_time_percentiles"] = percentile_report(ct)
        results["iteration_times_mean"] = (
            float(np.mean(extract_iteration_times(master_log)))
            if extract_iteration_times(master_log) else None)
    if os.path.exists(eval_log):
        times, losses, precs, steps = extract_times_losses_precision(eval_log)
        if times:
            results["eval"] = {
                "times": times, "losses": losses, "precisions": precs,
                "steps": steps, "final_precision": precs[-1],
            }
    with open(os.path.join(outdir, "results.json"), "w") as f:
        json.dump(results, f, indent=2)
    with open(os.path.join(outdir, "results.txt"), "a") as f:
        f.write(f"{name} run_time={run_time:.1f}s max_step={results['max_step']}\n")
    print(json.dumps(results, indent=2))
    return results


def plot_figs(outdirs, dest):
    """The reference's full plot set (benchmark.py:165-263): step->loss,
    step->train-acc, step->sec/batch, time->validation-precision,
    time->validation-loss (both from the scraped out_evaluator lines) and
    the per-worker compute-time CDF (from the ELAPSED TIMES report)."""
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
    except ImportError:
        print("matplotlib unavailable; skipping plots")
        return
    fig, axes = plt.subplots(2, 3, figsize=(16, 9))
    (ax_sl, ax_sa, ax_sb), (ax_tp, ax_tl, ax_cdf) = axes
    for d in outdirs:
        master = os.path.join(d, "out_master")
        if not os.path.exists(master):
            continue
        recs = extract_step_records(master)
        name = os.path.basename(d)
        if recs:
            steps = [r[0] for r in recs]
            ax_sl.plot(steps, [r[1] for r in recs], label=name)
            ax_sa.plot(steps, [r[2] for r in recs], label=name)
            ax_sb.plot(steps, [r[4] for r in recs], label=name)
        ev = os.path.join(d, "out_evaluator")
        if os.path.exists(ev):
            times, losses, precs, _steps = extract_times_losses_precision(ev)
            if times:
                ax_tp.plot(times, precs, marker="o", label=name)
                ax_tl.plot(times, losses, marker="o", label=name)
        ct = extract_compute_times(master)
        if ct:
            ts = np.sort(np.array([t for (t, _, _) in ct]))
            ax_cdf.plot(ts, np.arange(1, len(ts) + 1) / len(ts), label=name)
    ax_sl.set_xlabel("step"); ax_sl.set_ylabel("loss")
    ax_sa.set_xlabel("step"); ax_sa.set_ylabel("train acc")
    ax_sb.set_xlabel("step"); ax_sb.set_ylabel("sec/batch")
    ax_tp.set_xlabel("time (s)"); ax_tp.set_ylabel("validation precision @ 1")
    ax_tl.set_xlabel("time (s)"); ax_tl.set_ylabel("validation loss")
    ax_cdf.set_xlabel("per-worker compute time (s)"); ax_cdf.set_ylabel("CDF")
    for row in axes:
        for ax in row:
            ax.legend(fontsize=6)
    os.makedirs(dest, exist_ok=True)
    fig.savefig(os.path.join(dest, "sweep.png"), dpi=120)
    print(f"wrote {dest}/sweep.png")


def main(argv):
    if len(argv) < 2:
        print(__doc__)
        return 1
    cmd = argv[0]
    if cmd == "run":
        run_cfg(argv[1])
    elif cmd == "use_dir":
        outs = []
        for fn in sorted(os.listdir(argv[1])):
            if fn.endswith(".json"):
                r = run_cfg(os.path.join(argv[1], fn))
                outs.append(os.path.join(ROOT, "bench_out", r["name"]))
        plot_figs(outs, os.path.join(ROOT, "bench_out", "plots"))
    elif cmd == "scrape":
        print(json.dumps({
            "max_step": extract_steps(argv[1]),
            "percentiles": percentile_report(extract_compute_times(argv[1])),
        }, indent=2))
    else:
        print(__doc__)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
