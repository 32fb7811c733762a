"""Benchmark-harness tests: scraper contracts + a 2-worker CPU cfg run."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(ROOT, "tools"))

import benchmark as bm  # noqa: E402


def test_scrape_step_records(tmp_path):
    log = tmp_path / "out_master"
    log.write_text(
        "INFO:dmnist:Worker 0: 2026-09-13 10:00:00: step 5, loss = 2.301234, "
        "train_acc = 0.101562, test_acc = 0.000000(1234.5 examples/sec; "
        "0.052  sec/batch)\n"
        "garbage line\n"
        "INFO:dmnist:Worker 0: 2026-09-13 10:00:01: step 6, loss = 2.200000, "
        "train_acc = 0.200000, test_acc = 0.000000(2000.0 examples/sec; "
        "0.032  sec/batch)\n")
    assert bm.extract_steps(log) == 6
    recs = bm.extract_step_records(log)
    assert len(recs) == 2
    assert recs[0] == (5, 2.301234, 0.101562, 1234.5, 0.052)


def test_scrape_cdf_lines(tmp_path):
    log = tmp_path / "out_master"
    elapsed = [(0.01, 0, 21), (0.02, 1, 21), (0.015, 0, 22), (0.025, 1, 22)]
    log.write_text(
        f"INFO:dmnist:ELAPSED TIMES {elapsed}\n"
        "INFO:dmnist:ITERATION TIMES [0.011, 0.012]\n")
    ct = bm.extract_compute_times(log)
    assert ct == elapsed
    it = bm.extract_iteration_times(log)
    assert it == [0.011, 0.012]
    rep = bm.percentile_report(ct)
    assert rep["max"] == 0.025
    assert 0.0 < rep["mean"] < 0.03


def test_scrape_eval_lines(tmp_path):
    log = tmp_path / "out_eval"
    log.write_text(
        "Succesfully loaded model from /tmp/t/model.ckpt-30 at step=30.\n"
        "Num examples: 10000  Precision @ 1: 0.901200 Loss: 0.310000 "
        "Time: 12.500000\n")
    t, l, p, s = bm.extract_times_losses_precision(log)
    assert t == [12.5] and l == [0.31] and p == [0.9012] and s == [30]


@pytest.mark.timeout(300)
def test_run_cfg_cpu_2worker(tmp_path):
    cfg = {
        "name": "test_w2",
        "workers": 2,
        "timeout_s": 240,
        "flags": {"model": "mlp", "batch_size": 16, "max_steps": 4,
                  "synthetic_data": True, "device": "cpu",
                  "train_dir": str(tmp_path / "td"),
                  "save_interval_secs": 100000},
    }
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    res = bm.run_cfg(str(cfg_path), outdir=str(tmp_path / "out"))
    assert res["returncode"] == 0
    assert res["max_step"] >= 3
    assert "examples_per_sec_per_worker_p50" in res
    assert os.path.exists(tmp_path / "out" / "results.json")


@pytest.mark.timeout(300)
def test_run_cfg_with_evaluator_and_plots(tmp_path):
    """VERDICT round-1 task 7: run_cfg launches the evaluator per sweep
    point, scrapes its 'Precision @ 1' lines into results, and plot_figs
    emits the reference's full plot set (time->precision + time-CDF)."""
    cfg = {
        "name": "test_eval_w1",
        "workers": 1,
        "timeout_s": 240,
        "evaluator": True,
        "eval_interval_secs": 1,
        "flags": {"model": "mlp", "batch_size": 16, "max_steps": 25,
                  "synthetic_data": True, "device": "cpu",
                  "worker_times_cdf_method": True,
                  "train_dir": str(tmp_path / "td"),
                  "save_interval_secs": 1},
    }
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    out = tmp_path / "out"
    res = bm.run_cfg(str(cfg_path), outdir=str(out))
    assert res["returncode"] == 0
    assert os.path.exists(out / "out_evaluator")
    assert "eval" in res, open(out / "out_evaluator").read()[-1500:]
    assert res["eval"]["precisions"], res["eval"]
    assert 0.0 <= res["eval"]["final_precision"] <= 1.0
    bm.plot_figs([str(out)], str(tmp_path / "plots"))
    assert os.path.exists(tmp_path / "plots" / "sweep.png")
