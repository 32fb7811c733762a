#!/usr/bin/env python3
"""Reference-compatible training entry point
(cf. /root/reference/src/mnist_distributed_train.py).

Single machine, one process per MI355X GPU:
    torchrun --standalone --nproc-per-node 8 src/mnist_distributed_train.py \
        --synthetic_data --max_steps 1000
Legacy flags (--job_name/--ps_hosts/--worker_hosts/--task_id) are accepted;
'ps' roles exit immediately — there is no parameter server on an xGMI node.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distributedmnist_amd.engine.train import train_main  # noqa: E402
from distributedmnist_amd.utils.flags import build_train_parser  # noqa: E402


def main(argv=None):
    flags = build_train_parser().parse_args(argv)
    if flags.job_name == "ps":
        print("No parameter server in the MI355X design (gradients are "
              "all-reduced over xGMI); ps role exits.")
        return 0
    train_main(flags)
    return 0


if __name__ == "__main__":
    sys.exit(main())
