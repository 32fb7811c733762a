"""Trainer-level tests of the interval / CDF / k-of-N mode plumbing (CPU)."""

import logging

import torch

from distributedmnist_amd.engine.train import Trainer, make_dataset
from distributedmnist_amd.utils.flags import build_train_parser


def mk(tmp_path, *extra):
    argv = ["--synthetic_data", "--train_dir", str(tmp_path / "t"),
            "--batch_size", "8", "--model", "mlp", "--device", "cpu",
            "--save_interval_secs", "100000"] + list(extra)
    return build_train_parser().parse_args(argv)


def test_interval_mode_defers_updates(tmp_path):
    # huge interval: no update ever fires within the run
    flags = mk(tmp_path, "--interval_method", "--interval_ms", "1e9")
    t = Trainer(flags)
    assert t.mode == "interval"
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    w0 = t.fp.flat_master.clone()
    for _ in range(5):
        x, y = ds.next_batch(8)
        applied, *_ = t.train_step(*t.to_device(x, y))
        assert not applied
    torch.testing.assert_close(t.fp.flat_master, w0)
    # zero interval: fires every step, averaging the accumulated grads
    flags2 = mk(tmp_path, "--interval_method", "--interval_ms", "0")
    t2 = Trainer(flags2)
    ds2 = make_dataset(flags2, 0, 1, t2.device, t2.compute_dtype)
    w0 = t2.fp.flat_master.clone()
    x, y = ds2.next_batch(8)
    applied, *_ = t2.train_step(*t2.to_device(x, y))
    assert applied
    assert not torch.equal(t2.fp.flat_master, w0)


def test_cdf_mode_emits_scraper_lines(tmp_path, caplog):
    flags = mk(tmp_path, "--worker_times_cdf_method", "--max_steps", "60")
    t = Trainer(flags)
    assert t.mode == "cdf"
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    with caplog.at_level(logging.INFO):
        for _ in range(60):
            x, y = ds.next_batch(8)
            t.train_step(*t.to_device(x, y))
    text = caplog.text
    assert "ELAPSED TIMES" in text
    assert "ITERATION TIMES" in text
    # percentile stats available (benchmark.py:97-111 shape)
    stats = t.engine.compute_time_percentiles()
    assert stats and stats["p95"] >= stats["p80"] >= 0


def test_mode_selection_from_flags(tmp_path):
    assert Trainer(mk(tmp_path)).mode == "full_sync"
    assert Trainer(mk(tmp_path, "--interval_method")).mode == "interval"
    assert Trainer(mk(tmp_path, "--worker_times_cdf_method")).mode == "cdf"
    # k_of_n needs K < world; at world=1 it stays full_sync
    assert Trainer(mk(tmp_path, "--num_replicas_to_aggregate", "1")).mode == "full_sync"
