"""Distributed training loop — re-expression of
/root/reference/src/distributed_train.py:109-416 for one 8xMI355X node.

The reference's per-step pipeline (worker fwd/bwd -> gRPC gradient push -> PS
ConditionalAccumulator take_grad -> PS SGD apply -> token enqueue) becomes:

  fwd/bwd (HIP kernels) -> ONE flat all-reduce (RCCL over xGMI) -> identical
  fused SGD apply on every rank (grad_scale = 1/contributors folds the
  reference's take_grad average)

The token barrier (M5/M6) is subsumed by the collective's blocking
semantics; the Twisted startup barrier (M9) is init_process_group + an
initial broadcast of the flat parameters.  The reference's duplicate forward
pass (distributed_train.py:332-334 runs the same feed twice) is NOT copied —
loss/accuracy come out of the single training step.
"""

from __future__ import annotations

import datetime
import logging
import os
import time

import numpy as np
import torch
import torch.distributed as dist

from ..data import SyntheticDataSet
from ..models import build_model
from ..ops import functional as Fx
from ..parallel import FlatParams, StepTimer, SyncEngine
from .supervisor import Supervisor

log = logging.getLogger("dmnist.train")


def resolve_device(flag_device: str, local_rank: int = 0) -> torch.device:
    if flag_device not in ("auto", ""):
        return torch.device(flag_device)
    if torch.cuda.is_available():
        # modulo map so world_size > device_count still runs (exercising
        # multi-rank paths on a 1-GPU box with --backend gloo); the 8-GPU
        # launch maps 1:1
        return torch.device(f"cuda:{local_rank % torch.cuda.device_count()}")
    return torch.device("cpu")


def resolve_backend(flag_backend: str, device: torch.device) -> str:
    if flag_backend not in ("auto", ""):
        return flag_backend
    return "nccl" if device.type == "cuda" else "gloo"


def init_distributed(flags, device):
    """torchrun env rendezvous; returns (rank, world, local_rank)."""
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return 0, 1, 0
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if not dist.is_initialized():
        backend = resolve_backend(flags.backend, device)
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    return rank, world, local_rank


def lr_at(step: int, flags, num_examples: int, num_replicas: int) -> float:
    """Staircase exponential decay (distributed_train.py:143-156)."""
    num_batches_per_epoch = num_examples / flags.batch_size
    decay_steps = max(1, int(num_batches_per_epoch * flags.num_epochs_per_decay
                             / max(1, num_replicas)))
    return (flags.initial_learning_rate *
            flags.learning_rate_decay_factor ** (step // decay_steps))


class Trainer:
    """Owns model + flat buffers + sync engine; one step() per iteration."""

    def __init__(self, flags, device=None, rank=0, world=1, local_rank=0):
        self.flags = flags
        self.rank, self.world = rank, world
        self.device = device if device is not None else resolve_device(flags.device, local_rank)
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        if flags.compute_dtype == "auto":
            self.compute_dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        else:
            self.compute_dtype = dict(fp32=torch.float32, bf16=torch.bfloat16)[flags.compute_dtype]
        self.model = build_model(flags.model, seed=flags.seed,
                                 compute_dtype=self.compute_dtype).to(self.device)
        self.fp = FlatParams(self.model, device=self.device,
                             compute_dtype=self.compute_dtype)
        mode = "full_sync"
        if flags.interval_method:
            mode = "interval"
        elif flags.worker_times_cdf_method:
            mode = "cdf"
        elif 0 < flags.num_replicas_to_aggregate < world:
            mode = "k_of_n"
        self.mode = mode
        self.engine = SyncEngine(
            self.fp.flat_grad, mode=mode,
            replicas_to_aggregate=flags.num_replicas_to_aggregate,
            interval_ms=flags.interval_ms,
            straggler_timeout_ms=(flags.straggler_timeout_ms or None),
            rank=rank, world_size=world,
            wire_dtype=(torch.bfloat16
                        if getattr(flags, "grad_dtype", "fp32") == "bf16"
                        else None))
        self.timer = StepTimer(self.device)
        # drop-connect placement: pre-aggregation per-rank masks are the
        # reference's estimator (distributed_train.py:194-203);
        # --drop_connect_post keeps the round-1 shared post-aggregation form
        self._dc_pre = (flags.drop_connect
                        and not getattr(flags, "drop_connect_post", False))
        self.is_chief = rank == 0
        self.step = 0
        self.num_contributors = world
        self._num_examples = 60000  # overwritten by train(); LR-schedule default
        self._graph = None
        self._graph_tried = False
        self._eager_fused = None
        self._eager_fused_tried = False
        self._step_dev = None
        self.flat_momentum = (torch.zeros_like(self.fp.flat_master)
                              if flags.momentum > 0 else None)
        # parameter-init parity across ranks (broadcast once)
        if world > 1 and dist.is_initialized():
            dist.broadcast(self.fp.flat_master, src=0)
            self.fp.sync_shadow()
            dist.barrier()

    # ------------------------------------------------------------------
    def to_device(self, images, labels):
        if isinstance(images, np.ndarray):
            images = torch.from_numpy(images.copy())
            labels = torch.from_numpy(np.ascontiguousarray(labels))
        images = images.to(device=self.device, dtype=self.compute_dtype,
                           non_blocking=True)
        labels = labels.to(device=self.device, non_blocking=True)
        return images, labels

    def get_graph(self, images):
        """hipGraph-captured step when eligible (GPU, full_sync, no
        injection/timing instrumentation); None -> eager path."""
        if self._graph is not None:
            return self._graph
        if self._graph_tried:
            return None
        self._graph_tried = True
        flags = self.flags
        # cdf at world=1 is graphable too: no collective can pollute the
        # per-worker timing, so hipEvents around the replay measure exactly
        # what the eager path measures (at world>1 cdf stays eager — the
        # straggler CDF must be COMPUTE time, not compute+comm)
        graphable_mode = (self.mode == "full_sync"
                          or (self.mode == "cdf" and self.world == 1))
        eligible = (self.device.type == "cuda" and graphable_mode
                    and self.engine.timeout_s is None
                    and flags.inject_slow_rank < 0
                    and getattr(flags, "hip_graph", "auto") != "off")
        if eligible:
            from .graphstep import try_graph
            # capture scope: world > 1 captures the COMPUTE only (split) so
            # no collective is ever inside a capture — a backend that
            # refuses capture invalidates it unrecoverably (segfault; see
            # docs/ROUND1_NOTES.md).  --hip_graph full (or the
            # DMNIST_FORCE_GRAPH diagnostic env) forces whole-step capture
            # including the RCCL all-reduce.
            force_full = (getattr(flags, "hip_graph", "auto") == "full"
                          or bool(os.environ.get("DMNIST_FORCE_GRAPH")))
            split = (self.world > 1 and dist.is_initialized()
                     and not force_full)
            self._graph = try_graph(self, tuple(images.shape), split=split)
        if self.world > 1 and dist.is_initialized():
            # ALL ranks must agree on graph-vs-eager: the graphed fused step
            # issues a bucketed 2-collective sequence, eager issues one —
            # a mixed fleet would deadlock the communicator.
            ok = torch.tensor(
                [1.0 if self._graph is not None else 0.0],
                device=self.device if self.device.type == "cuda" else "cpu")
            # watchdog: a rank that died during capture would leave the
            # others hung here forever — bound the wait and fail loudly
            import datetime as _dt
            work = dist.all_reduce(ok, op=dist.ReduceOp.MIN, async_op=True)
            try:
                done = work.wait(timeout=_dt.timedelta(seconds=180))
            except Exception as e:  # noqa: BLE001 — backend-specific timeout
                raise RuntimeError(
                    "graph-agreement all-reduce failed/timed out: a peer "
                    "rank likely died during hipGraph capture") from e
            if done is False:
                raise RuntimeError(
                    "graph-agreement all-reduce timed out after 180s: a "
                    "peer rank likely died during hipGraph capture")
            if float(ok.item()) < 0.5 and self._graph is not None:
                log.warning("hipGraph disabled: another rank failed capture")
                self._graph = None
        return self._graph

    def graph_or_eager_step(self, images, labels):
        """Preferred step entry: graph replay when available."""
        g = self.get_graph(images)
        if g is not None:
            if self.mode == "cdf":
                step = self.step
                self.engine.step_begin(step)
                e0, e1 = self._cdf_event_pair()
                e0.record()
                loss, acc = g.run(images, labels)
                e1.record()
                self._cdf_steps.append(step)
                if len(self._cdf_steps) >= self.engine.cdf_log_every:
                    self._flush_cdf()
                return True, loss, acc, 0.0
            loss, acc = g.run(images, labels)
            return True, loss, acc, 0.0
        return self.train_step(images, labels)

    # -- batched CDF timing for the graph path -------------------------
    def _cdf_event_pair(self):
        if not hasattr(self, "_cdf_events"):
            n = self.engine.cdf_log_every
            self._cdf_events = [(torch.cuda.Event(enable_timing=True),
                                 torch.cuda.Event(enable_timing=True))
                                for _ in range(n)]
            self._cdf_steps = []
        return self._cdf_events[len(self._cdf_steps)]

    def _flush_cdf(self):
        """One synchronize for the whole window, then one gather+report
        (identical content to the per-step form; ~70 us/step cheaper)."""
        if not getattr(self, "_cdf_steps", None):
            return
        n = len(self._cdf_steps)
        self._cdf_events[n - 1][1].synchronize()
        times = [self._cdf_events[i][0].elapsed_time(
                     self._cdf_events[i][1]) / 1000.0 for i in range(n)]
        self.engine.record_cdf_batch(self._cdf_steps, times)
        self._cdf_steps = []

    def _get_eager_fused(self):
        """FusedLeNetStep for the eager path (non-graphable modes), or None
        (CPU / non-LeNet / fp32 / DMNIST_EAGER_FUSED=0)."""
        if self._eager_fused_tried:
            return self._eager_fused
        self._eager_fused_tried = True
        from ..models import LeNet5
        if (self.device.type == "cuda"
                and isinstance(self.model, LeNet5)
                and self.model.shadows and self.model.shadows_T
                and getattr(self.flags, "fused_step", "auto") != "off"
                and os.environ.get("DMNIST_EAGER_FUSED", "1") != "0"):
            from .fused_step import FusedLeNetStep
            f = FusedLeNetStep(self)
            f.overlap_allreduce = False  # SyncEngine owns the collective
            self._eager_fused = f
            self._step_dev = torch.zeros(1, dtype=torch.int64,
                                         device=self.device)
        return self._eager_fused

    @property
    def _needs_step_timing(self) -> bool:
        """Host-synchronizing per-step timers are only needed for the
        instrumented/straggler modes; the full-sync hot path stays async."""
        return (self.mode in ("k_of_n", "cdf")
                or self.engine.timeout_s is not None)

    def train_step(self, images, labels):
        """One synchronous step.

        Returns (applied, loss, acc, compute_time); loss/acc are 0-dim
        DEVICE tensors on the hot path (no forced host sync — call float()
        only when logging)."""
        flags = self.flags
        self.model.set_step(self.step)
        self.engine.step_begin(self.step)
        timed = self._needs_step_timing
        if timed:
            self.timer.start()
        if flags.inject_slow_rank == self.rank and flags.inject_slow_ms > 0:
            time.sleep(flags.inject_slow_ms / 1000.0)
        self.fp.zero_grad()
        fused = self._get_eager_fused()
        if fused is not None:
            # hand-scheduled fused step in EAGER mode too: the instrumented
            # modes (cdf / k_of_n / interval / straggler) get the same
            # kernel path as the captured graph; the collective stays with
            # SyncEngine.reduce below (fused.overlap_allreduce disabled)
            self._step_dev.fill_(self.step)
            loss, acc = fused(images, labels, self._step_dev)
        else:
            logits = self.model(images, train=True)
            loss, acc = self.model.loss_and_accuracy(logits, labels)
            loss.backward()
            self.fp.fix_grad_views()
        if self._dc_pre:
            Fx.grad_mask(self.fp.flat_grad, flags.drop_connect_probability,
                         flags.seed, self.step, self.rank)
        compute_time = self.timer.stop() if timed else 0.0
        applied, grad, contributors = self.engine.reduce(self.step, compute_time)
        if applied:
            self._apply_update(grad, contributors)
        self.step += 1
        return applied, loss.detach(), acc.detach(), compute_time

    def _apply_update(self, grad, contributors):
        """Fused SGD apply of one aggregated gradient.  Interval mode keys
        the LR staircase on the aggregation count (the reference's
        global_step increments once per take_grad apply, not per worker
        step); every other mode keys it on the local step as before."""
        flags = self.flags
        if self.mode == "interval":
            lr_step = max(0, self.engine.generation - 1)
        else:
            lr_step = self.step
        lr = lr_at(lr_step, flags, self._num_examples,
                   max(1, self.engine.K))
        Fx.sgd_step(self.fp.flat_master, grad, lr,
                    grad_scale=1.0 / max(1, contributors),
                    drop_connect_keep=(flags.drop_connect_probability
                                       if (flags.drop_connect
                                           and not self._dc_pre) else None),
                    seed=flags.seed, offset=lr_step,
                    shadow=self.fp.flat_shadow,
                    momentum=self.flat_momentum, mu=flags.momentum)
        self.fp.refresh_transposes()
        self.num_contributors = contributors

    # ------------------------------------------------------------------
    def train(self, dataset, max_steps=None):
        flags = self.flags
        self._num_examples = dataset.num_examples
        max_steps = max_steps if max_steps is not None else flags.max_steps
        sv = Supervisor(flags.train_dir, flags.save_interval_secs,
                        is_chief=self.is_chief)
        restored = Supervisor.restore(flags.train_dir) if os.path.isdir(flags.train_dir) else None
        if restored is not None:
            step0, payload = restored
            self.fp.load_flat(payload["flat_master"])
            if (self.flat_momentum is not None
                    and payload.get("flat_momentum") is not None):
                self.flat_momentum.copy_(
                    payload["flat_momentum"].to(self.device))
            self.step = step0
            if self.is_chief:
                log.info("Restored checkpoint at step %d", step0)
        time_acc_list = []
        begin_time = time.time()
        writer = None
        if flags.should_summarize and self.is_chief:
            from ..utils.tbwriter import make_writer
            writer = make_writer(flags.train_dir)
        next_summary_time = time.time() + flags.save_summaries_secs
        while self.step < max_steps:
            start_time = time.time()
            images, labels = dataset.next_batch(flags.batch_size)
            images, labels = self.to_device(images, labels)
            if flags.timeline_logging and 5 <= self.step <= 7:
                # chrome trace per step (reference FULL_TRACE timelines,
                # distributed_train.py:317-358) via torch.profiler/kineto
                from torch.profiler import ProfilerActivity, profile
                acts = [ProfilerActivity.CPU]
                if self.device.type == "cuda":
                    acts.append(ProfilerActivity.CUDA)
                with profile(activities=acts) as prof:
                    applied, loss_v, acc_v, _ct = self.train_step(images, labels)
                torch.cuda.synchronize() if self.device.type == "cuda" else None
                os.makedirs(flags.train_dir, exist_ok=True)
                prof.export_chrome_trace(os.path.join(
                    flags.train_dir,
                    f"worker={self.rank}_timeline_iter={self.step}.json"))
                finish_time = time.time()
            else:
                applied, loss_v, acc_v, _ct = self.graph_or_eager_step(images, labels)
                finish_time = time.time()
            loss_v, acc_v = float(loss_v), float(acc_v)
            duration = finish_time - start_time
            examples_per_sec = flags.batch_size / duration
            # per-step line: scraper contract (benchmark.py:31 'step (\d+),')
            log.info("Worker %d: %s: step %d, loss = %f, train_acc = %f, "
                     "test_acc = %f(%.1f examples/sec; %.3f  sec/batch)",
                     self.rank, datetime.datetime.now(), self.step, loss_v,
                     acc_v, 0.0, examples_per_sec, duration)
            time_acc_list.append((finish_time, acc_v, 0.0, loss_v))
            if self.step % flags.save_results_period == 0:
                path = os.path.join(flags.train_dir,
                                    f"worker{self.rank}_time_acc.npy")
                try:
                    os.makedirs(flags.train_dir, exist_ok=True)
                    np.save(path, np.array(time_acc_list, dtype=np.float64))
                except OSError:
                    pass
            if writer is not None and time.time() > next_summary_time:
                writer.add_scalar("Train Loss", loss_v, self.step)
                writer.add_scalar("Train Accuracy", acc_v, self.step)
                writer.add_scalar("Examples/sec", examples_per_sec, self.step)
                next_summary_time += flags.save_summaries_secs
            sv.maybe_save(self.step, self.checkpoint_payload)
        if self.mode == "interval":
            # drain: agree on the total generation count out-of-band, post
            # anything still owed, apply every remaining aggregation — all
            # ranks leave with identical parameters
            for grad, contributors in self.engine.finalize_interval():
                self._apply_update(grad, contributors)
        if self.mode == "cdf":
            self._flush_cdf()  # tail of the last (partial) timing window
        if writer is not None:
            writer.close()
        if self.is_chief:
            log.info("Elapsed Time: %f", time.time() - begin_time)
            sv.save(self.step, self.checkpoint_payload())
        if self.world > 1 and dist.is_initialized():
            dist.barrier()
        return time_acc_list

    def checkpoint_payload(self):
        return {
            "flat_master": self.fp.flat_master.detach().cpu().clone(),
            "flat_momentum": (self.flat_momentum.detach().cpu().clone()
                              if self.flat_momentum is not None else None),
            "model_state": {k: v.cpu() for k, v in
                            self.fp.state_dict_params().items()},
            "model": self.flags.model,
            "seed": self.flags.seed,
        }


def make_dataset(flags, rank: int, world: int, device, dtype):
    if flags.synthetic_data:
        pool = max(4 * flags.batch_size, 8192)
        return SyntheticDataSet(pool_size=pool, device=device, dtype=dtype,
                                seed=flags.seed + rank)
    from ..data import load_mnist
    ds = load_mnist(flags.data_dir, fake_data=flags.fake_data,
                    worker_id=rank, n_workers=world,
                    shard=not flags.no_shard, seed=flags.seed + rank)
    train = ds.train
    if flags.subset and not flags.fake_data:
        # reference --subset: train on the first N examples
        from ..data import DataSet
        train = DataSet(train.images[:flags.subset],
                        train.labels[:flags.subset], shard=False,
                        seed=flags.seed + rank)
    return train


def train_main(flags):
    """CLI entry (src/mnist_distributed_train.py)."""
    from ..utils.logging import setup
    device = resolve_device(flags.device, int(os.environ.get("LOCAL_RANK", "0")))
    rank, world, local_rank = init_distributed(flags, device)
    setup(rank)
    device = resolve_device(flags.device, local_rank)
    trainer = Trainer(flags, device=device, rank=rank, world=world,
                      local_rank=local_rank)
    ds = make_dataset(flags, rank, world, trainer.device, trainer.compute_dtype)
    out = trainer.train(ds)
    if world > 1 and dist.is_initialized():
        dist.destroy_process_group()
    return out
