#!/usr/bin/env python3
"""Reference-compatible evaluator entry point
(cf. /root/reference/src/mnist_eval.py): wipes/creates eval_dir, then polls
checkpoint_dir evaluating the validation set (which, as in the reference, is
the MNIST test set — mnist_data.py:200-201)."""

import os
import shutil
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distributedmnist_amd.data import load_mnist, SyntheticDataSet  # noqa: E402
from distributedmnist_amd.engine.evaluate import evaluate  # noqa: E402
from distributedmnist_amd.utils.flags import build_eval_parser  # noqa: E402


def main(argv=None):
    flags = build_eval_parser().parse_args(argv)
    if os.path.exists(flags.eval_dir):
        shutil.rmtree(flags.eval_dir, ignore_errors=True)
    os.makedirs(flags.eval_dir, exist_ok=True)
    if flags.synthetic_data:
        import torch

        class _SynthEval:
            def __init__(self):
                ds = SyntheticDataSet(pool_size=10000, device="cpu",
                                      dtype=torch.float32, seed=4242)
                self.images = ds._images
                self.labels = ds._labels
                self.num_examples = 10000
        dataset = _SynthEval()
    else:
        dataset = load_mnist(flags.data_dir, fake_data=flags.fake_data,
                             shard=False).validation
    from distributedmnist_amd.utils.tbwriter import make_writer
    writer = make_writer(flags.eval_dir)
    evaluate(dataset, flags, writer=writer)
    return 0


if __name__ == "__main__":
    sys.exit(main())
