#!/usr/bin/env python3
"""Helper for tests/test_rccl_multigpu.py::test_param_parity_across_gpus:
train 20 sync steps at world=2 over the default (nccl/RCCL on GPU) backend,
then dump each rank's flat_master for a cross-rank parity check."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    out_dir = sys.argv[1]
    from distributedmnist_amd.engine.train import (Trainer, init_distributed,
                                                   make_dataset,
                                                   resolve_device)
    from distributedmnist_amd.utils.flags import build_train_parser
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--batch_size", "256", "--max_steps", "20",
         "--save_interval_secs", "100000",
         "--train_dir", os.path.join(out_dir, "train")])
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    device = resolve_device(flags.device, local_rank)
    rank, world, local_rank = init_distributed(flags, device)
    t = Trainer(flags, device=device, rank=rank, world=world,
                local_rank=local_rank)
    ds = make_dataset(flags, rank, world, t.device, t.compute_dtype)
    for _ in range(20):
        x, y = ds.next_batch(flags.batch_size)
        t.graph_or_eager_step(*t.to_device(x, y))
    if t.device.type == "cuda":
        torch.cuda.synchronize()
    torch.save(t.fp.flat_master.detach().cpu(),
               os.path.join(out_dir, f"rank{rank}.pt"))
    if world > 1:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
