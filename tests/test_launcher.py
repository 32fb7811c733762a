"""Local launcher (tools/launch.py) tests — tf_ec2.py subcommand parity."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LAUNCH = os.path.join(ROOT, "tools", "launch.py")


@pytest.mark.timeout(240)
def test_run_tf_single_worker(tmp_path):
    cfg = {"name": "launch_test", "workers": 1,
           "flags": {"model": "mlp", "batch_size": 8, "max_steps": 3,
                     "synthetic_data": True, "device": "cpu",
                     "train_dir": str(tmp_path / "td"),
                     "save_interval_secs": 100000}}
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    env = dict(os.environ, DMNIST_RUN_DIR=str(tmp_path / "runs"))
    out = subprocess.run([sys.executable, LAUNCH, "run_tf", str(cfg_path)],
                         capture_output=True, text=True, timeout=180, env=env)
    assert out.returncode == 0, out.stdout + out.stderr
    log = tmp_path / "runs" / "launch_test" / "out_master"
    assert log.exists() and "step 2," in log.read_text()


def test_noop_subcommands():
    out = subprocess.run([sys.executable, LAUNCH, "shutdown"],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 0 and "no-op" in out.stdout


def test_run_command():
    out = subprocess.run([sys.executable, LAUNCH, "run_command", "echo hi_launcher"],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 0
