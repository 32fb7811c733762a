"""Single-process end-to-end engine tests (BASELINE config 1 class)."""

import os

import numpy as np
import torch

from distributedmnist_amd.engine.supervisor import Supervisor
from distributedmnist_amd.engine.train import Trainer, lr_at, make_dataset
from distributedmnist_amd.utils.flags import build_train_parser


def mkflags(tmp_path, *extra):
    argv = ["--synthetic_data", "--train_dir", str(tmp_path / "train"),
            "--batch_size", "32", "--max_steps", "6", "--model", "mlp",
            "--device", "cpu", "--save_results_period", "2"] + list(extra)
    return build_train_parser().parse_args(argv)


def test_trainer_runs_and_loss_decreases(tmp_path):
    flags = mkflags(tmp_path, "--model", "lenet", "--max_steps", "12",
                    "--batch_size", "64", "--initial_learning_rate", "0.05")
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    hist = t.train(ds)
    assert len(hist) == 12
    losses = [h[3] for h in hist]
    assert all(np.isfinite(losses))
    # training on a small synthetic pool should reduce loss
    assert np.mean(losses[-4:]) < np.mean(losses[:4])


def test_trainer_writes_checkpoint_and_npy(tmp_path):
    flags = mkflags(tmp_path)
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    t.train(ds)
    latest = Supervisor.latest_checkpoint(flags.train_dir)
    assert latest is not None and latest[0] == 6
    assert os.path.exists(os.path.join(flags.train_dir, "worker0_time_acc.npy"))
    arr = np.load(os.path.join(flags.train_dir, "worker0_time_acc.npy"))
    assert arr.shape[1] == 4  # (finish_time, train_acc, test_acc, loss)


def test_trainer_resume(tmp_path):
    flags = mkflags(tmp_path)
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    t.train(ds)
    w_after6 = t.fp.flat_master.clone()
    # new trainer restores from the checkpoint and continues
    flags2 = mkflags(tmp_path)
    flags2.max_steps = 10
    t2 = Trainer(flags2)
    ds2 = make_dataset(flags2, 0, 1, t2.device, t2.compute_dtype)
    torch.testing.assert_close(t2.fp.flat_master, t2.fp.flat_master)
    t2.train(ds2)
    assert t2.step == 10
    # restored start point equals saved weights
    restored = Supervisor.restore(flags.train_dir)
    assert restored[0] == 10


def test_lr_schedule_staircase():
    flags = build_train_parser().parse_args(
        ["--batch_size", "64", "--initial_learning_rate", "0.01",
         "--num_epochs_per_decay", "1", "--learning_rate_decay_factor", "0.5"])
    # decay_steps = (60000/64)*1/1 = 937
    assert lr_at(0, flags, 60000, 1) == 0.01
    assert lr_at(936, flags, 60000, 1) == 0.01
    assert abs(lr_at(937, flags, 60000, 1) - 0.005) < 1e-12
    assert abs(lr_at(2 * 937, flags, 60000, 1) - 0.0025) < 1e-12


def test_straggler_timeout_world1_skips_update(tmp_path):
    flags = mkflags(tmp_path, "--straggler_timeout_ms", "0.000001",
                    "--inject_slow_rank", "0", "--inject_slow_ms", "5")
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    w0 = t.fp.flat_master.clone()
    x, y = ds.next_batch(32)
    applied, *_ = t.train_step(*t.to_device(x, y))
    assert not applied
    torch.testing.assert_close(t.fp.flat_master, w0)


def test_drop_connect_trains(tmp_path):
    flags = mkflags(tmp_path, "--drop_connect")
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    hist = t.train(ds)
    assert all(np.isfinite([h[3] for h in hist]))
