"""Flag-interaction matrix: every DP mode crossed with drop-connect
placement and the bf16 gradient wire, world_size=2 over gloo.  Each cell
trains 6 steps end-to-end through Trainer.train() and checks rank-0/1
parameter agreement for the synchronous modes — interaction bugs (e.g. a
mode that skips the mask, a wire buffer aliasing a mode's flag element)
don't show up in the single-feature tests."""

import os
import sys

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from mp_utils import free_port, init_pg, spawn_collect  # noqa: E402,F401

CASES = [
    # (extra argv, params_must_match)
    ([], True),
    (["--drop_connect"], True),                        # pre-agg masks
    (["--drop_connect", "--drop_connect_post"], True),
    (["--grad_dtype", "bf16"], True),
    (["--drop_connect", "--grad_dtype", "bf16"], True),
    (["--num_replicas_to_aggregate", "1"], True),
    (["--num_replicas_to_aggregate", "1", "--grad_dtype", "bf16"], True),
    (["--worker_times_cdf_method"], True),
    (["--worker_times_cdf_method", "--drop_connect"], True),
    (["--interval_method", "--interval_ms", "0"], True),
    (["--interval_method", "--interval_ms", "20", "--drop_connect"], True),
    (["--straggler_timeout_ms", "10000"], True),
    (["--momentum", "0.9"], True),
    (["--momentum", "0.9", "--drop_connect", "--grad_dtype", "bf16"], True),
]


def _run(rank, world, port, extra, q):
    init_pg(rank, world, port)
    from distributedmnist_amd.engine.train import Trainer, make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--train_dir", f"/tmp/dmnist_matrix_{port}",
         "--batch_size", "16", "--max_steps", "6", "--model", "mlp",
         "--device", "cpu", "--save_interval_secs", "100000"] + extra)
    t = Trainer(flags, rank=rank, world=world, local_rank=rank)
    ds = make_dataset(flags, rank, world, t.device, t.compute_dtype)
    t.train(ds)
    q.put((rank, float(t.fp.flat_master.sum()),
           t.fp.flat_master[:4].tolist()))
    dist.destroy_process_group()


@pytest.mark.timeout(240)
@pytest.mark.parametrize("extra,must_match",
                         CASES, ids=[" ".join(c[0]) or "default"
                                     for c in CASES])
def test_flag_matrix_world2(extra, must_match):
    res = spawn_collect(_run, 2, args=(extra,))
    res = {r: (v[0], tuple(v[1])) for r, v in res.items()}
    if must_match:
        assert res[0] == res[1], (extra, res)


def _run_wire_traj(rank, world, port, grad_dtype, q):
    init_pg(rank, world, port)
    from distributedmnist_amd.engine.train import Trainer, make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--train_dir", f"/tmp/dmnist_wt_{port}",
         "--batch_size", "16", "--max_steps", "20", "--model", "mlp",
         "--device", "cpu", "--save_interval_secs", "100000",
         "--grad_dtype", grad_dtype])
    t = Trainer(flags, rank=rank, world=world, local_rank=rank)
    ds = make_dataset(flags, rank, world, t.device, t.compute_dtype)
    t.train(ds)
    q.put((rank, t.fp.flat_master.clone()))
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_bf16_wire_trajectory_tracks_fp32():
    """20-step 2-rank trajectories with fp32 vs bf16 gradient wire must
    stay close (bf16 rounds each reduce; fp32 master updates otherwise
    identical) — the numerics guard for --grad_dtype bf16."""
    import torch as _torch

    def run(dtype):
        res = spawn_collect(_run_wire_traj, 2, args=(dtype,))
        assert _torch.equal(res[0][0], res[1][0])
        return res[0][0]

    w32 = run("fp32")
    w16 = run("bf16")
    rel = float((w32 - w16).norm() / w32.norm())
    assert rel < 0.01, f"bf16-wire trajectory diverged: rel {rel}"
