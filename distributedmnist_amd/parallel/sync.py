"""Gradient sync engine — RCCL-over-xGMI re-expression of the reference's
four DP synchronization flavors:

  full_sync  DP-1: SyncReplicasOptimizer K=N  -> one flat all-reduce/step
  k_of_n     DP-1: K<N backup workers        -> contribute-or-zero +
             count-renormalize (slowest N-K ranks' grads dropped per step,
             decided from an all-gathered per-rank compute time)
  interval   DP-3: wall-clock timer updates (sync_replicas_optimizer_
             modified.py:208-215) -> local grad accumulation, rank-0 timer
             broadcast, all-reduce at interval boundaries only
  cdf        DP-2: per-worker token barrier + compute-time CDF collection
             (timeout_manager.py:48-70) -> the all-reduce IS the barrier;
             per-rank step timers all-gathered, 'ELAPSED TIMES'/'ITERATION
             TIMES' log lines kept scraper-compatible (benchmark.py:66,140)

Straggler timeout (DP-4, vestigial in the reference): straggler_timeout_ms
drops a rank's own contribution when its compute time exceeds the deadline
(always-contribute-but-zero, renormalized by the surviving count — no
communicator reconstruction needed).

The flat gradient buffer is extended by one trailing element (the
contribution flag) so the contributor count rides in the SAME all-reduce —
no extra collective on the critical path for k_of_n/timeout modes.
"""

from __future__ import annotations

import logging
import time

import torch
import torch.distributed as dist

log = logging.getLogger("dmnist.sync")

MODES = ("full_sync", "k_of_n", "interval", "cdf")


class SyncEngine:
    def __init__(self, flat_grad: torch.Tensor, mode: str = "full_sync",
                 replicas_to_aggregate: int | None = None,
                 interval_ms: float | None = None,
                 straggler_timeout_ms: float | None = None,
                 group=None, rank: int = 0, world_size: int = 1,
                 cdf_log_every: int = 50, cdf_start_tracking: int = 20):
        if mode not in MODES:
            raise ValueError(f"mode {mode!r} not in {MODES}")
        self.mode = mode
        self.flat_grad = flat_grad
        self.group = group
        self.rank = rank
        self.world = world_size
        self.K = replicas_to_aggregate if replicas_to_aggregate and replicas_to_aggregate > 0 else world_size
        self.K = min(self.K, world_size)
        self.interval_s = (interval_ms or 0.0) / 1000.0
        self.timeout_s = (straggler_timeout_ms / 1000.0) if straggler_timeout_ms else None
        dev = flat_grad.device
        # flag element rides in the same buffer: [grad..., flag]
        self._buf = torch.empty(flat_grad.numel() + 1, dtype=flat_grad.dtype, device=dev)
        self._accum = None
        self._accum_count = 0
        if mode == "interval":
            self._accum = torch.zeros_like(flat_grad)
            self._interval_t0 = None
        self._ctrl = torch.zeros(1, dtype=torch.float32, device=dev)
        # cdf instrumentation
        self.cdf_log_every = cdf_log_every
        self.cdf_start_tracking = cdf_start_tracking
        self._compute_times = []      # [(elapsed, worker, iteration), ...]
        self._iter_start_times = []

    @property
    def distributed(self) -> bool:
        return self.world > 1 and dist.is_initialized()

    # ------------------------------------------------------------------
    def step_begin(self, step: int):
        self._t_start = time.time()
        if self.mode == "cdf":
            # (step, t_start) pairs: list position is NOT the step number
            # after a checkpoint restore (training resumes at step0 > 0)
            self._iter_start_times.append((step, self._t_start))

    def _all_gather_times(self, my_time: float, step: int):
        """Collect per-rank compute times (cdf instrumentation / k_of_n
        ranking). One small all_gather_into_tensor + ONE host sync (a
        per-element .item() loop costs `world` separate syncs per step —
        at 8 ranks that becomes the step time)."""
        if not self.distributed:
            return [my_time]
        t = torch.tensor([my_time], dtype=torch.float64,
                         device=self.flat_grad.device)
        out = torch.zeros(self.world, dtype=torch.float64, device=t.device)
        dist.all_gather_into_tensor(out, t, group=self.group)
        return out.tolist()

    # ------------------------------------------------------------------
    def reduce(self, step: int, compute_time_s: float | None = None):
        """Aggregate gradients for this step.

        Returns (apply_update: bool, grad_tensor, contributors: int).
        grad_tensor is the SUM over contributors; the caller folds the
        1/contributors into the fused SGD apply.
        """
        if compute_time_s is None:
            compute_time_s = time.time() - getattr(self, "_t_start", time.time())

        if self.mode == "interval":
            return self._reduce_interval(step)

        # hot path: pure synchronous all-reduce, no staging copies, no
        # host round-trips — the collective IS the barrier
        if self.mode == "full_sync" and self.timeout_s is None:
            if self.distributed:
                dist.all_reduce(self.flat_grad, op=dist.ReduceOp.SUM,
                                group=self.group)
            return True, self.flat_grad, self.world

        contribute = True
        if self.timeout_s is not None and compute_time_s > self.timeout_s:
            contribute = False
            log.info("Worker %d: step %d compute time %.3fs exceeded straggler "
                     "timeout %.3fs; dropping contribution", self.rank, step,
                     compute_time_s, self.timeout_s)
        if self.mode == "k_of_n" and self.K < self.world:
            times = self._all_gather_times(compute_time_s, step)
            order = sorted(range(self.world), key=lambda r: (times[r], r))
            if self.rank not in order[:self.K]:
                contribute = False

        if self.mode == "cdf":
            times = self._all_gather_times(compute_time_s, step)
            for w, tm in enumerate(times):
                self._compute_times.append((tm, w, step))
            self._maybe_log_cdf(step)

        n = self.flat_grad.numel()
        if not self.distributed:
            # world=1: a dropped contribution means no update this step
            return contribute, self.flat_grad, 1

        buf = self._buf
        if contribute:
            buf[:n].copy_(self.flat_grad)
            buf[n] = 1.0
        else:
            buf[:n].zero_()
            buf[n] = 0.0
        dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=self.group)
        contributors = max(1, int(round(float(buf[n].item()))))
        self.flat_grad.copy_(buf[:n])
        return True, self.flat_grad, contributors

    # ------------------------------------------------------------------
    def _reduce_interval(self, step: int):
        """DP-3: accumulate locally; all-reduce + apply only when rank 0's
        wall-clock timer says the interval elapsed (broadcast each step)."""
        self._accum += self.flat_grad
        self._accum_count += 1
        if self._interval_t0 is None:
            self._interval_t0 = time.time()
        fire = 0.0
        if self.rank == 0 and (time.time() - self._interval_t0) >= self.interval_s:
            fire = 1.0
        if self.distributed:
            self._ctrl[0] = fire
            dist.broadcast(self._ctrl, src=0, group=self.group)
            fire = float(self._ctrl.item())
        if fire < 0.5:
            return False, None, 0
        # interval fired: average everything accumulated everywhere
        n = self.flat_grad.numel()
        buf = self._buf
        buf[:n].copy_(self._accum)
        buf[n] = float(self._accum_count)
        if self.distributed:
            dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=self.group)
        contributors = max(1, int(round(float(buf[n].item()))))
        self.flat_grad.copy_(buf[:n])
        self._accum.zero_()
        self._accum_count = 0
        self._interval_t0 = time.time()
        if self.rank == 0:
            log.info("Interval update fired at step %d (averaged %d grads)",
                     step, contributors)
        return True, self.flat_grad, contributors

    # ------------------------------------------------------------------
    def _maybe_log_cdf(self, step: int):
        """Scraper-compatible CDF report (timeout_manager.py:63-70 format,
        consumed by benchmark.py extract_compute_times/iteration_times)."""
        if self.rank != 0 or step <= self.cdf_start_tracking:
            return
        if step % self.cdf_log_every != 0:
            return
        elapsed = sorted((t, w, i) for (t, w, i) in self._compute_times
                         if i > self.cdf_start_tracking)
        starts = [t for (s, t) in self._iter_start_times
                  if s > self.cdf_start_tracking]
        iter_times = [starts[i + 1] - starts[i] for i in range(len(starts) - 1)]
        log.info("ELAPSED TIMES %s", str(elapsed))
        log.info("ITERATION TIMES %s", str(iter_times))

    def compute_time_percentiles(self):
        """Percentile stats over collected per-rank compute times
        (benchmark.py:97-111 shape)."""
        import numpy as np
        if not self._compute_times:
            return {}
        times = np.array([t for (t, _, _) in self._compute_times])
        return {
            "std": float(times.std()),
            "max": float(times.max()),
            "p80": float(np.percentile(times, 80)),
            "p90": float(np.percentile(times, 90)),
            "p95": float(np.percentile(times, 95)),
            "p99": float(np.percentile(times, 99)),
            "mean": float(times.mean()),
        }
