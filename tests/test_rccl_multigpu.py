"""RCCL multi-GPU gate (VERDICT round-1 task 2).

Every multi-rank artifact in round 1 ran over gloo (2 ranks sharing one
GPU).  These tests are the SCALE-day gate: they auto-skip at
device_count < 2 and, the first time the driver's round-end `pytest -m gpu`
lands on a multi-GPU box, they exercise the REAL RCCL path with no builder
action — world=2 over the nccl(=RCCL) backend: bench.py end-to-end, the
two-graph split capture with the eager fc all-reduce on a comm stream, all
four DP modes, and the bf16 wire-dtype reduce.

Design under test: sync engine (parallel/sync.py) re-expressing
/root/reference/src/sync_replicas_optimizer_modified.py:237-429 as bucketed
RCCL all-reduces over xGMI.
"""

import json
import os
import re
import subprocess
import sys

import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(torch.cuda.device_count() < 2,
                       reason="needs >= 2 GPUs (RCCL refuses duplicate "
                              "devices; gloo covers the 1-GPU case)"),
]


def _torchrun(args, port, env_extra=None, timeout=420):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    env.pop("DMNIST_BACKEND", None)  # force the nccl/RCCL default
    if env_extra:
        env.update(env_extra)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port)] + args,
        capture_output=True, text=True, timeout=timeout, cwd=ROOT, env=env)
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    return out


def _last_json_line(stdout):
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout[-2000:]}")


@pytest.mark.timeout(600)
def test_bench_2gpu_rccl_two_graph_split():
    """The flagship bench over actual RCCL: two-graph split capture must
    engage (graph A replay, eager fc-slice all-reduce on a comm stream
    overlapping graph B, conv-slice all-reduce, device-arg tail)."""
    out = _torchrun([os.path.join(ROOT, "bench.py"), "--gpus", "2",
                     "--steps", "40", "--warmup", "10"], port=29451)
    j = _last_json_line(out.stdout)
    assert j["n_gpus"] == 2 and j["config"]["parallelism"] == "dp2"
    assert j["value"] > 0
    combined = out.stdout + out.stderr
    assert "running eager" not in combined, combined[-1500:]


@pytest.mark.timeout(600)
def test_bench_2gpu_rccl_bf16_wire():
    out = _torchrun([os.path.join(ROOT, "bench.py"), "--gpus", "2",
                     "--steps", "20", "--warmup", "5",
                     "--grad_dtype", "bf16"], port=29452)
    j = _last_json_line(out.stdout)
    assert j["config"]["grad_dtype"] == "bf16" and j["value"] > 0


@pytest.mark.timeout(900)
@pytest.mark.parametrize("mode_args", [
    ([],                                              "full_sync"),
    (["--num_replicas_to_aggregate", "1"],            "k_of_n"),
    (["--interval_method", "--interval_ms", "50"],    "interval"),
    (["--worker_times_cdf_method"],                   "cdf"),
], ids=["full_sync", "k_of_n", "interval", "cdf"])
def test_all_modes_2gpu_rccl(mode_args, tmp_path):
    """All four DP modes of the reference over real RCCL, 30 steps each;
    the per-step scraper line must appear and training must reach the end."""
    extra, name = mode_args
    port = 29460 + hash(name) % 20
    out = _torchrun(
        [os.path.join(ROOT, "src", "mnist_distributed_train.py"),
         "--synthetic_data", "--batch_size", "256", "--max_steps", "30",
         "--save_interval_secs", "100000",
         "--train_dir", str(tmp_path / name)] + extra,
        port=port, timeout=800)
    txt = out.stdout + out.stderr
    assert re.search(r"step 29, loss = [0-9.]+", txt), txt[-2000:]


@pytest.mark.timeout(600)
def test_hip_graph_full_capture_over_rccl(tmp_path):
    """--hip_graph full captures the WHOLE step including the RCCL
    all-reduce (ROUND2 candidate the 1-GPU boxes could never exercise);
    30 steps at world=2 must complete with the capture engaged."""
    out = _torchrun(
        [os.path.join(ROOT, "src", "mnist_distributed_train.py"),
         "--synthetic_data", "--batch_size", "256", "--max_steps", "30",
         "--save_interval_secs", "100000", "--hip_graph", "full",
         "--train_dir", str(tmp_path / "full")],
        port=29456)
    txt = out.stdout + out.stderr
    assert re.search(r"step 29, loss = [0-9.]+", txt), txt[-2000:]
    assert "running eager" not in txt, txt[-1500:]


@pytest.mark.timeout(600)
def test_param_parity_across_gpus_after_training(tmp_path):
    """Bitwise-policy check per SURVEY section 7.2 slice 2: after N sync
    steps over RCCL both ranks hold (near-)identical parameters — the
    all-reduce delivers identical sums, so only fp32-atomic nondeterminism
    inside a rank's own backward may differ, and the checksum files must
    agree to that tolerance."""
    helper = os.path.join(ROOT, "tools", "rccl_parity_probe.py")
    out = _torchrun([helper, str(tmp_path)], port=29455)
    a = torch.load(os.path.join(tmp_path, "rank0.pt"), weights_only=True)
    b = torch.load(os.path.join(tmp_path, "rank1.pt"), weights_only=True)
    torch.testing.assert_close(a, b, rtol=0, atol=1e-5)
