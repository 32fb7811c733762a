"""bench.py driver-contract tests (CPU): single-proc and the torchrun
launch pattern the driver uses for N>1."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json_line(stdout):
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout}")


@pytest.mark.timeout(300)
def test_bench_single_process():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch_size", "64"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr
    j = _last_json_line(out.stdout)
    assert j["n_gpus"] == 1 and j["steps"] == 3 and j["warmup"] == 1
    assert j["unit"] == "images/sec" and j["value"] > 0
    assert j["scaling"] == "weak" and j["higher_is_better"] is True
    assert j["config"]["global_batch"] == 64
    assert j["config"]["parallelism"] == "dp1"
    assert j["data"] == "synthetic"


def test_bench_json_schema_complete():
    """Every field the driver contract names must be present with the
    right type/value domain (a missing or mistyped field invalidates the
    round's BENCH record)."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch_size", "32"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json_line(out.stdout)
    assert isinstance(j["metric"], str) and j["metric"]
    assert isinstance(j["value"], (int, float)) and j["value"] > 0
    assert j["unit"] == "images/sec"
    assert j["n_gpus"] == 1 and j["steps"] == 3 and j["warmup"] == 1
    assert isinstance(j["ms_per_step"], (int, float)) and j["ms_per_step"] > 0
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["vs_baseline"] is None  # reference publishes no numbers
    assert j["dtype"] in ("bf16", "fp32")
    assert j["data"] == "synthetic"
    cfg = j["config"]
    assert cfg["model"] == "lenet"
    assert cfg["global_batch"] == 32
    assert cfg["seq_len"] is None
    assert cfg["parallelism"] == "dp1"
    for k in ("p50_ms_per_step", "p95_ms_per_step", "p99_ms_per_step"):
        assert isinstance(cfg[k], (int, float)) and cfg[k] > 0
    # whole-job aggregate identity: value = global_batch * steps / elapsed
    approx = cfg["global_batch"] / (j["ms_per_step"] / 1000.0)
    assert abs(j["value"] - approx) < 0.01 * approx


@pytest.mark.timeout(300)
def test_bench_mode_is_real_cdf():
    """VERDICT round-1 task 1: --mode must map onto the trainer flags, not
    just label the JSON.  A cdf bench demonstrably runs the cdf engine and
    the JSON reports the mode that ran (trainer.mode, not the CLI arg)."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "60",
         "--warmup", "1", "--batch_size", "32", "--mode", "cdf"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json_line(out.stdout)
    assert j["config"]["mode"] == "cdf"
    assert j["config"]["mode_requested"] == "cdf"


@pytest.mark.timeout(300)
def test_bench_mode_interval_and_straggler():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "5",
         "--warmup", "1", "--batch_size", "32", "--mode", "interval",
         "--interval_ms", "10", "--straggler", "0", "--straggler_ms", "1"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json_line(out.stdout)
    assert j["config"]["mode"] == "interval"


@pytest.mark.timeout(300)
def test_bench_mode_k_of_n_honest_at_world1():
    """k_of_n degenerates to full_sync at world=1 — the JSON must say what
    actually ran (mode) alongside what was asked (mode_requested)."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch_size", "32", "--mode", "k_of_n"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    j = _last_json_line(out.stdout)
    assert j["config"]["mode_requested"] == "k_of_n"
    assert j["config"]["mode"] == "full_sync"


@pytest.mark.timeout(600)
def test_bench_torchrun_2proc_k_of_n():
    """2-rank k_of_n bench: the mode engages for real (K=1 of 2)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29418",
         os.path.join(ROOT, "bench.py"), "--gpus", "2", "--steps", "3",
         "--warmup", "1", "--batch_size", "32", "--mode", "k_of_n",
         "--replicas_to_aggregate", "1"],
        capture_output=True, text=True, timeout=540, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-3000:]
    j = _last_json_line(out.stdout)
    assert j["config"]["mode"] == "k_of_n"


@pytest.mark.timeout(600)
def test_bench_torchrun_2proc():
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29417",
         os.path.join(ROOT, "bench.py"), "--gpus", "2", "--steps", "3",
         "--warmup", "1", "--batch_size", "32"],
        capture_output=True, text=True, timeout=540, cwd=ROOT, env=env)
    assert out.returncode == 0, out.stderr[-3000:]
    j = _last_json_line(out.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["global_batch"] == 64
    assert j["config"]["parallelism"] == "dp2"


@pytest.mark.gpu
def test_bench_two_rank_gloo_on_one_gpu():
    """Multi-rank GPU path end-to-end on a single device: 2 ranks share
    cuda:0 over gloo (RCCL refuses duplicate devices), exercising the init
    broadcast, the fused step, the bucketed all-reduce and the
    graph-eligibility gate (gloo must fall back to eager cleanly)."""
    env = dict(os.environ)
    env.update({"DMNIST_BACKEND": "gloo", "MASTER_ADDR": "127.0.0.1"})
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29381", os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "8", "--warmup", "2"],
        capture_output=True, text=True, timeout=280, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0
    # the two-graph split capture must actually engage: an eager fallback
    # would still produce a valid JSON line but lose the graph replay and
    # the overlapped fc all-reduce at scale
    combined = out.stdout + out.stderr
    assert "running eager" not in combined, combined[-1500:]


@pytest.mark.gpu
def test_two_graph_split_matches_eager_numerics(tmp_path):
    """The two-graph split capture must train the same trajectory as the
    eager multi-rank path (within fp32-atomic noise): same seeds, same
    synthetic data, 30 steps, 2 ranks sharing one GPU over gloo."""
    import re

    def run(port, extra):
        env = dict(os.environ)
        env["DMNIST_BACKEND"] = "gloo"
        out = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", str(port),
             os.path.join(ROOT, "src", "mnist_distributed_train.py"),
             "--synthetic_data", "--backend", "gloo", "--batch_size", "128",
             "--max_steps", "30", "--save_interval_secs", "100000",
             "--train_dir", str(tmp_path / f"t{port}")] + extra,
            capture_output=True, text=True, timeout=280, env=env, cwd=ROOT)
        assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
        txt = out.stdout + out.stderr
        losses = re.findall(r"Worker 0: .*step 29, loss = ([0-9.]+)", txt)
        assert losses, txt[-1500:]
        return float(losses[0]), txt

    loss_g, txt_g = run(29392, [])
    assert "running eager" not in txt_g  # the capture must engage
    loss_e, _ = run(29393, ["--hip_graph", "off"])
    assert abs(loss_g - loss_e) < 0.05 * max(1.0, abs(loss_e)), \
        (loss_g, loss_e)


@pytest.mark.gpu
def test_graph_drop_connect_matches_eager(tmp_path):
    """Per-rank pre-aggregation drop-connect INSIDE the two-graph capture
    (grad_mask keyed on step_dev) must track the eager path's trajectory:
    same philox stream (seed, step, rank), so losses match within
    fp32-atomic noise."""
    import re

    def run(port, extra):
        env = dict(os.environ)
        env["DMNIST_BACKEND"] = "gloo"
        out = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", str(port),
             os.path.join(ROOT, "src", "mnist_distributed_train.py"),
             "--synthetic_data", "--backend", "gloo", "--batch_size", "128",
             "--max_steps", "25", "--save_interval_secs", "100000",
             "--drop_connect",
             "--train_dir", str(tmp_path / f"dc{port}")] + extra,
            capture_output=True, text=True, timeout=280, env=env, cwd=ROOT)
        assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
        txt = out.stdout + out.stderr
        losses = re.findall(r"Worker 0: .*step 24, loss = ([0-9.]+)", txt)
        assert losses, txt[-1500:]
        return float(losses[0]), txt

    loss_g, txt_g = run(29394, [])
    assert "running eager" not in txt_g
    loss_e, _ = run(29395, ["--hip_graph", "off"])
    assert abs(loss_g - loss_e) < 0.05 * max(1.0, abs(loss_e)), \
        (loss_g, loss_e)
