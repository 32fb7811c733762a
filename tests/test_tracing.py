"""Timeline logging (torch.profiler chrome traces) + summaries."""

import json
import os

from distributedmnist_amd.engine.train import Trainer, make_dataset
from distributedmnist_amd.utils.flags import build_train_parser


def test_timeline_logging_writes_chrome_trace(tmp_path):
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--train_dir", str(tmp_path / "t"),
         "--batch_size", "8", "--max_steps", "8", "--model", "mlp",
         "--device", "cpu", "--timeline_logging",
         "--save_interval_secs", "100000"])
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    t.train(ds)
    traces = [f for f in os.listdir(flags.train_dir)
              if f.startswith("worker=0_timeline_iter=")]
    assert traces, os.listdir(flags.train_dir)
    with open(os.path.join(flags.train_dir, traces[0])) as f:
        data = json.load(f)
    assert "traceEvents" in data
