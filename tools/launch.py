#!/usr/bin/env python3
"""Local cluster launcher — the single-node counterpart of
/root/reference/tools/tf_ec2.py (C8 in SURVEY.md section 2.1).

The reference launched an EC2 fleet (spot requests, SSH command templating,
NFS, scp downloads).  On one 8xMI355X node the "fleet" is N ranks under
torch.distributed.run plus an evaluator process; the same subcommand surface
is kept where it still means something:

  run_tf <cfg.json>      launch training (torchrun) + evaluator, wait, save logs
  kill_all_python        kill processes started BY THIS TOOL (pidfile-scoped;
                         never pattern-kills)
  run_command "<cmd>"    run a shell command locally (tf_ec2.py:744-768)
  download_file <f> <d>  copy a file from the run dir (tf_ec2.py:651-742)
  download_outdir <d>    copy the whole run dir
  shutdown / launch      no-ops with an explanation (no fleet to manage)

Cfg files are the JSON documents of cfg/ (tools/benchmark.py format).
"""

from __future__ import annotations

import json
import os
import shutil
import signal
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
RUN_DIR = os.environ.get("DMNIST_RUN_DIR", "/tmp/dmnist_runs")
PIDFILE = os.path.join(RUN_DIR, "launch.pids")


def _record_pid(pid: int):
    os.makedirs(RUN_DIR, exist_ok=True)
    with open(PIDFILE, "a") as f:
        f.write(f"{pid}\n")


def run_tf(cfg_path: str, wait: bool = True):
    with open(cfg_path) as f:
        cfg = json.load(f)
    name = cfg.get("name") or os.path.splitext(os.path.basename(cfg_path))[0]
    outdir = os.path.join(RUN_DIR, name)
    os.makedirs(outdir, exist_ok=True)
    nproc = int(cfg.get("workers", 1))
    flags = dict(cfg.get("flags", {}))
    train_dir = flags.setdefault("train_dir", os.path.join(outdir, "train_dir"))
    argv = []
    for k, v in flags.items():
        if isinstance(v, bool):
            if v:
                argv.append(f"--{k}")
        else:
            argv += [f"--{k}", str(v)]
    entry = os.path.join(ROOT, "src", "mnist_distributed_train.py")
    if nproc > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--standalone",
               "--local-addr", "127.0.0.1", f"--nproc-per-node={nproc}",
               entry] + argv
    else:
        cmd = [sys.executable, entry] + argv
    master_log = open(os.path.join(outdir, "out_master"), "w")
    train_proc = subprocess.Popen(cmd, stdout=master_log,
                                  stderr=subprocess.STDOUT, cwd=ROOT,
                                  start_new_session=True)
    _record_pid(train_proc.pid)
    eval_proc = None
    if cfg.get("evaluator", False):
        eval_log = open(os.path.join(outdir, "out_evaluator"), "w")
        ecmd = [sys.executable, os.path.join(ROOT, "src", "mnist_eval.py"),
                "--checkpoint_dir", train_dir,
                "--eval_dir", os.path.join(outdir, "eval_dir"),
                "--synthetic_data" if flags.get("synthetic_data") else
                "--fake_data",
                "--model", str(flags.get("model", "lenet"))]
        eval_proc = subprocess.Popen(ecmd, stdout=eval_log,
                                     stderr=subprocess.STDOUT, cwd=ROOT,
                                     start_new_session=True)
        _record_pid(eval_proc.pid)
    print(f"launched {name}: train pid {train_proc.pid}"
          + (f", eval pid {eval_proc.pid}" if eval_proc else ""))
    if wait:
        rc = train_proc.wait()
        if eval_proc is not None:
            try:
                os.killpg(eval_proc.pid, signal.SIGTERM)
            except ProcessLookupError:
                pass
        print(f"{name} finished rc={rc}; logs in {outdir}")
        return rc
    return 0


def kill_all_python():
    """Kill ONLY processes this tool launched (exact pids from the pidfile —
    never a pattern kill)."""
    if not os.path.exists(PIDFILE):
        print("nothing launched by this tool")
        return
    with open(PIDFILE) as f:
        pids = [int(line) for line in f if line.strip()]
    for pid in pids:
        try:
            os.killpg(pid, signal.SIGTERM)
            print(f"killed pgid {pid}")
        except ProcessLookupError:
            pass
    os.remove(PIDFILE)


def main(argv):
    if not argv:
        print(__doc__)
        return 1
    cmd = argv[0]
    if cmd == "run_tf":
        return run_tf(argv[1])
    if cmd == "kill_all_python":
        kill_all_python()
    elif cmd == "run_command":
        return subprocess.run(argv[1], shell=True, cwd=ROOT).returncode
    elif cmd == "download_file":
        src = os.path.join(RUN_DIR, argv[1])
        shutil.copy(src, argv[2])
        print(os.path.join(argv[2], os.path.basename(src)))
    elif cmd == "download_outdir":
        for name in os.listdir(RUN_DIR):
            d = os.path.join(RUN_DIR, name)
            if os.path.isdir(d):
                shutil.copytree(d, os.path.join(argv[1], name),
                                dirs_exist_ok=True)
    elif cmd in ("shutdown", "launch", "clean_launch"):
        print(f"'{cmd}' is a no-op: the cluster is this node's 8 GPUs "
              "(the reference managed an EC2 fleet here, tf_ec2.py:237-323)")
    else:
        print(__doc__)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
