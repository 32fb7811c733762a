"""Flag registry — argparse mirror of the reference's tf.app.flags surface
(/root/reference/src/distributed_train.py:36-99).

Reference flags are kept name-for-name so existing launch tooling maps over.
`ps_hosts` is accepted and ignored (there is no parameter server on an
8xMI355X node); `worker_hosts`/`task_id` are superseded by torchrun's
RANK/WORLD_SIZE env when present.  New MI355X-specific flags are grouped at
the bottom.
"""

from __future__ import annotations

import argparse
import os


def build_train_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="MI355X-native distributed MNIST training")
    # --- reference-compatible flags (distributed_train.py:36-99) ---
    p.add_argument("--worker_times_cdf_method", action="store_true",
                   help="collect per-worker compute-time CDFs (DP-2 mode)")
    p.add_argument("--interval_method", action="store_true",
                   help="wall-clock interval aggregation (DP-3 mode)")
    p.add_argument("--should_summarize", action="store_true",
                   help="write TensorBoard summaries on the chief")
    p.add_argument("--timeline_logging", action="store_true",
                   help="write per-step chrome traces (torch.profiler)")
    p.add_argument("--job_name", default="worker", choices=["ps", "worker"],
                   help="kept for CLI parity; 'ps' exits immediately (no PS on MI355X)")
    p.add_argument("--ps_hosts", default="", help="ignored (no parameter server)")
    p.add_argument("--worker_hosts", default="", help="comma-separated; superseded by torchrun env")
    p.add_argument("--train_dir", default="/tmp/mnist_train",
                   help="checkpoints + event logs directory")
    p.add_argument("--rpc_port", type=int, default=1235, help="vestigial (reference RPC mesh port)")
    p.add_argument("--save_results_period", type=int, default=20,
                   help="steps between worker time/accuracy npy dumps")
    p.add_argument("--max_steps", type=int, default=1000)
    p.add_argument("--drop_connect", action="store_true")
    p.add_argument("--batch_size", type=int, default=64, help="per-worker batch size")
    p.add_argument("--subset", type=int, default=0, help="train on a subset of N examples (0=all)")
    p.add_argument("--log_device_placement", action="store_true", help="vestigial")
    p.add_argument("--task_id", type=int, default=None,
                   help="worker rank; defaults to env RANK")
    p.add_argument("--num_replicas_to_aggregate", type=int, default=-1,
                   help="K of K-of-N aggregation; -1 => K = world size")
    p.add_argument("--save_interval_secs", type=int, default=20)
    p.add_argument("--save_summaries_secs", type=int, default=300)
    p.add_argument("--initial_learning_rate", type=float, default=0.01)
    p.add_argument("--momentum", type=float, default=0.0,
                   help="SGD momentum mu (0 = reference GradientDescent parity)")
    p.add_argument("--num_epochs_per_decay", type=float, default=1.0)
    p.add_argument("--learning_rate_decay_factor", type=float, default=0.95)
    p.add_argument("--drop_connect_probability", type=float, default=0.9,
                   help="keep probability of the per-element gradient mask")
    p.add_argument("--interval_ms", type=float, default=1000.0,
                   help="interval-method aggregation period")
    # --- MI355X-native additions ---
    p.add_argument("--model", default="lenet", choices=["lenet", "mlp"])
    p.add_argument("--synthetic_data", action="store_true",
                   help="device-resident synthetic MNIST-shaped data")
    p.add_argument("--fake_data", action="store_true", help="reference fake-data mode")
    p.add_argument("--data_dir", default="data", help="MNIST idx-gz directory")
    p.add_argument("--no_shard", action="store_true",
                   help="reference parity: every worker sees the full set")
    p.add_argument("--compute_dtype", default="auto", choices=["auto", "fp32", "bf16"],
                   help="auto => bf16 on GPU, fp32 on CPU")
    p.add_argument("--straggler_timeout_ms", type=float, default=0.0,
                   help=">0: drop own gradient when compute exceeds this (DP-4)")
    p.add_argument("--inject_slow_rank", type=int, default=-1,
                   help="rank to slow down for straggler testing")
    p.add_argument("--inject_slow_ms", type=float, default=0.0,
                   help="sleep this long per step on the injected rank")
    p.add_argument("--seed", type=int, default=66478)
    p.add_argument("--device", default="auto", help="auto|cpu|cuda")
    p.add_argument("--hip_graph", default="auto",
                   choices=["auto", "off", "full"],
                   help="capture the training step in a hipGraph (GPU full-sync)")
    p.add_argument("--fused_step", default="auto", choices=["auto", "off"],
                   help="hand-scheduled two-stream LeNet step inside the graph")
    p.add_argument("--backend", default="auto", help="auto|nccl|gloo")
    p.add_argument("--grad_dtype", default="fp32", choices=["fp32", "bf16"],
                   help="all-reduce wire dtype: bf16 halves the xGMI payload "
                        "(6.65->3.33 MB); the fp32 master update is unchanged")
    p.add_argument("--drop_connect_post", action="store_true",
                   help="legacy round-1 form: one shared mask on the "
                        "AGGREGATED gradient (default is the reference's "
                        "per-worker pre-aggregation masks, "
                        "distributed_train.py:194-203)")
    return p


def build_eval_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="MI355X-native MNIST evaluator")
    # reference flags (nn_eval.py:36-44, mnist_eval.py)
    p.add_argument("--eval_dir", default="/tmp/mnist_eval")
    p.add_argument("--checkpoint_dir", default="/tmp/mnist_train")
    p.add_argument("--eval_interval_secs", type=int, default=30)
    p.add_argument("--run_once", action="store_true")
    p.add_argument("--fake_data", action="store_true")
    p.add_argument("--synthetic_data", action="store_true")
    p.add_argument("--data_dir", default="data")
    p.add_argument("--model", default="lenet", choices=["lenet", "mlp"])
    p.add_argument("--device", default="auto")
    p.add_argument("--max_evals", type=int, default=0, help="0 = run forever")
    return p


def resolve_rank_world(flags) -> tuple[int, int]:
    """torchrun env wins; else task_id/worker_hosts; else single process."""
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        return int(os.environ["RANK"]), int(os.environ["WORLD_SIZE"])
    world = len(flags.worker_hosts.split(",")) if flags.worker_hosts else 1
    rank = flags.task_id if flags.task_id is not None else 0
    return rank, world
