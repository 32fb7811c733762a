"""Gradient sync engine — RCCL-over-xGMI re-expression of the reference's
four DP synchronization flavors:

  full_sync  DP-1: SyncReplicasOptimizer K=N  -> one flat all-reduce/step
  k_of_n     DP-1: K<N backup workers        -> contribute-or-zero +
             count-renormalize (slowest N-K ranks' grads dropped per step,
             decided from an all-gathered per-rank compute time)
  interval   DP-3: wall-clock timer updates (sync_replicas_optimizer_
             modified.py:208-215) -> FREE-RUNNING workers: local grad
             accumulation, fire times derived from a once-broadcast shared
             clock origin (one node = one clock), aggregation posted as an
             ASYNC all-reduce on a ring of staging buffers and applied on
             completion — a slow rank delays when an update lands but never
             stalls another rank's compute, exactly the decoupling of the
             reference's free-running workers (distributed_train.py:271-288)
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
                 cdf_log_every: int = 50, cdf_start_tracking: int = 20,
                 wire_dtype: torch.dtype | None = None):
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
        # wire dtype: bf16 halves the all-reduce payload over xGMI
        # (6.65 -> 3.33 MB); contributor counts <= world are
        # exactly representable in bf16 so the flag element rides along fine
        self.wire_dtype = wire_dtype if wire_dtype != flat_grad.dtype else None
        # flag element rides in the same buffer: [grad..., flag]
        buf_dtype = self.wire_dtype or flat_grad.dtype
        self._buf = torch.empty(flat_grad.numel() + 1, dtype=buf_dtype, device=dev)
        self._accum = None
        self._accum_count = 0
        if mode == "interval":
            self._accum = torch.zeros_like(flat_grad)
            self._next_fire = None       # shared-clock fire schedule origin
            self._ring = [torch.empty_like(self._buf)
                          for _ in range(self.INTERVAL_RING)]
            self._free = list(range(self.INTERVAL_RING))
            self._pending = []           # [(gen, work, ring_idx), ...] FIFO
            self._deferred = []          # [(grad_clone, contributors), ...]
            self._gen_posted = 0
            self.generation = 0          # aggregations APPLIED (global-step
            #                              equivalent; drives the LR decay)
            self._gloo = None
            if world_size > 1 and dist.is_initialized():
                # CPU side-group for the out-of-band shutdown agreement —
                # never used on the per-step path
                self._gloo = dist.new_group(backend="gloo")
        self._ctrl = torch.zeros(1, dtype=torch.float32, device=dev)
        # cdf instrumentation
        self.cdf_log_every = cdf_log_every
        self.cdf_start_tracking = cdf_start_tracking
        self._compute_times = []      # [(elapsed, worker, iteration), ...]
        self._iter_start_times = []

    @property
    def distributed(self) -> bool:
        return self.world > 1 and dist.is_initialized()

    def wire_allreduce(self, tensor: torch.Tensor, start: int = 0):
        """SUM all-reduce of `tensor` (flat_grad or a slice of it at offset
        `start`) in the wire dtype.  With a bf16 wire the matching slice of
        the preallocated staging buffer is used, so disjoint slices can
        reduce concurrently on different streams (the graphed two-bucket
        fc/conv split) without aliasing."""
        if not self.distributed:
            return
        if self.wire_dtype is None:
            dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=self.group)
        else:
            w = self._buf[start:start + tensor.numel()]
            w.copy_(tensor)
            dist.all_reduce(w, op=dist.ReduceOp.SUM, group=self.group)
            tensor.copy_(w)

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

        if self.mode == "cdf":
            # instrumentation first (the per-rank time gather), then the
            # same fast reduce as full_sync below: cdf never drops a
            # contribution, so the flag-buffer staging path (two extra
            # full-bucket copies) is pure overhead for it
            times = self._all_gather_times(compute_time_s, step)
            for w, tm in enumerate(times):
                self._compute_times.append((tm, w, step))
            self._maybe_log_cdf(step)

        # hot path: pure synchronous all-reduce, no staging copies, no
        # host round-trips — the collective IS the barrier
        if self.mode in ("full_sync", "cdf") and self.timeout_s is None:
            if self.distributed:
                if self.wire_dtype is None:
                    dist.all_reduce(self.flat_grad, op=dist.ReduceOp.SUM,
                                    group=self.group)
                else:
                    n = self.flat_grad.numel()
                    wire = self._buf[:n]
                    wire.copy_(self.flat_grad)       # fp32 -> bf16 cast
                    dist.all_reduce(wire, op=dist.ReduceOp.SUM,
                                    group=self.group)
                    self.flat_grad.copy_(wire)       # bf16 -> fp32 cast
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
    INTERVAL_RING = 8   # max in-flight aggregations before backpressure

    def _reduce_interval(self, step: int):
        """DP-3 free-running: accumulate locally every step; when the shared
        fire schedule says an interval boundary passed, POST an async
        all-reduce of the accumulator and keep stepping.  The update is
        applied whenever the collective completes — there is no per-step
        collective, broadcast, or host sync coupling the ranks, so a slow
        rank cannot stall a fast one between firings (the coupling the
        reference's interval method exists to avoid,
        sync_replicas_optimizer_modified.py:208-215)."""
        self._accum += self.flat_grad
        self._accum_count += 1
        if self._next_fire is None:
            # one-time agreement on the fire-schedule origin: all ranks of
            # one node share a wall clock, so after this single broadcast
            # every rank computes every future fire time locally
            t = torch.tensor([time.time() + self.interval_s],
                             dtype=torch.float64, device=self._ctrl.device)
            if self.distributed:
                dist.broadcast(t, src=0, group=self.group)
            self._next_fire = float(t.item())
        now = time.time()
        if self.interval_s <= 0:
            # degenerate interval: fire-every-step, applied synchronously
            # (the per-step-averaging semantics the zero-interval cfg means)
            self._post_generation(step)
            return self._poll_apply(step, force=True)
        while now >= self._next_fire:
            # catch-up loop: a rank that slept through k boundaries owes k
            # generations (collectives match by posting order on the
            # communicator, so every rank must post every generation)
            self._post_generation(step)
            self._next_fire += self.interval_s
        return self._poll_apply(step)

    def _post_generation(self, step: int, block: bool = True) -> bool:
        """Post one aggregation generation.  block=False refuses instead of
        waiting when the ring is exhausted — REQUIRED during the shutdown
        phase-1 schedule servicing: a rank ahead of the schedule that
        hard-blocked there could wait on a generation its (already
        barrier-complete, no-longer-firing) peer will never post, while
        that peer blocks in the phase-2 MAX — mutual deadlock (found by the
        flag-matrix stress test).  In the training loop and in phase 2
        blocking is safe: every other rank is still servicing the schedule
        (training/phase-1) or known to be posting its owed generations."""
        n = self.flat_grad.numel()
        if not self._free:
            # completed-but-unapplied work still occupies slots (nothing
            # calls _poll_apply during shutdown phase 1): recycle it into
            # the deferred-apply queue before deciding to refuse/block
            self._recycle_completed()
        if not self._free:
            if not block:
                return False
            # backpressure: ring exhausted (this rank is > RING intervals
            # ahead of the slowest) — block on the oldest in-flight, move
            # its result to the deferred-apply queue, recycle its slot
            gen, work, idx = self._pending.pop(0)
            if work is not None:
                work.wait()
            buf_ = self._ring[idx]
            contributors = max(1, int(round(float(buf_[n].item()))))
            self._deferred.append((buf_[:n].clone(), contributors))
            self._free.append(idx)
        idx = self._free.pop()
        buf = self._ring[idx]
        buf[:n].copy_(self._accum)
        buf[n] = float(self._accum_count)
        work = None
        if self.distributed:
            work = dist.all_reduce(buf, op=dist.ReduceOp.SUM,
                                   group=self.group, async_op=True)
        self._pending.append((self._gen_posted, work, idx))
        self._gen_posted += 1
        self._accum.zero_()
        self._accum_count = 0
        return True

    def _recycle_completed(self):
        """Move every COMPLETED in-flight aggregation (oldest-first) into
        the deferred-apply queue, freeing its ring slot.  Keeps the ring
        live for ranks that are posting without applying (the shutdown
        phase-1 schedule servicing)."""
        n = self.flat_grad.numel()
        while self._pending:
            gen, work, idx = self._pending[0]
            if work is not None and not work.is_completed():
                break
            if work is not None:
                work.wait()  # ordering/stream registration; already done
            self._pending.pop(0)
            buf_ = self._ring[idx]
            contributors = max(1, int(round(float(buf_[n].item()))))
            self._deferred.append((buf_[:n].clone(), contributors))
            self._free.append(idx)

    def _poll_apply(self, step: int, force: bool = False):
        """Apply the oldest completed aggregation, if any (at most one per
        call; later completions surface on subsequent steps)."""
        if self._deferred:
            grad, contributors = self._deferred.pop(0)
            self.flat_grad.copy_(grad)
            self.generation += 1
            return True, self.flat_grad, contributors
        if not self._pending:
            return False, None, 0
        gen, work, idx = self._pending[0]
        if work is not None:
            if not force and not work.is_completed():
                return False, None, 0
            # NCCL: wait() inserts the stream dependency for the reads
            # below (host-blocking only for gloo); the collective itself
            # ran on the backend's internal stream, so pending aggregations
            # never serialize this rank's compute stream
            work.wait()
        self._pending.pop(0)
        n = self.flat_grad.numel()
        buf = self._ring[idx]
        contributors = max(1, int(round(float(buf[n].item()))))
        self.flat_grad.copy_(buf[:n])
        self._free.append(idx)
        self.generation += 1
        if self.rank == 0:
            log.info("Interval update %d applied at step %d (averaged %d "
                     "grads)", gen, step, contributors)
        return True, self.flat_grad, contributors

    def finalize_interval(self):
        """Shutdown drain (two-phase, over the CPU side-group, OUT of band
        of the training communicator):

        Phase 1 — while waiting for every rank to reach finalize (async
        gloo barrier), KEEP SERVICING THE FIRE SCHEDULE: a rank that
        stopped stepping must go on posting generations on time, or a
        slower rank still in its loop would block in ring backpressure
        waiting for posts that never come.

        Phase 2 — everyone is here and nobody is blocked on the training
        communicator, so a blocking MAX of the posted-generation counts is
        safe; post anything still owed, then yield every remaining
        aggregated gradient in order for the caller to apply."""
        if self.mode != "interval":
            return
        if self.distributed and self._gloo is not None:
            flag = torch.zeros(1, dtype=torch.int64)
            barrier = dist.all_reduce(flag, group=self._gloo, async_op=True)
            while not barrier.is_completed():
                now = time.time()
                while (self._next_fire is not None and self.interval_s > 0
                       and now >= self._next_fire):
                    # NON-blocking post: a full ring here must yield back
                    # to the barrier poll, never hard-block (deadlock
                    # against a peer already waiting in the phase-2 MAX)
                    if not self._post_generation(step=-1, block=False):
                        break
                    self._next_fire += self.interval_s
                time.sleep(0.002)
            g = torch.tensor([self._gen_posted], dtype=torch.int64)
            dist.all_reduce(g, op=dist.ReduceOp.MAX, group=self._gloo)
            target = int(g.item())
            while self._gen_posted < target:
                self._post_generation(step=-1)
        while self._pending or self._deferred:
            applied, grad, contributors = self._poll_apply(step=-1, force=True)
            if not applied:
                break
            yield grad, contributors
        # a later training run on this engine renegotiates its own fire
        # origin (a stale origin would owe one catch-up per elapsed
        # interval since THIS run ended)
        self._next_fire = None

    # ------------------------------------------------------------------
    def _maybe_log_cdf(self, step: int, force: bool = False):
        """Scraper-compatible CDF report (timeout_manager.py:63-70 format,
        consumed by benchmark.py extract_compute_times/iteration_times).
        force=True skips the every-N-steps gate (the batched path flushes
        exactly once per window, at whatever step ends it)."""
        if self.rank != 0 or step <= self.cdf_start_tracking:
            return
        if not force and step % self.cdf_log_every != 0:
            return
        if not log.isEnabledFor(logging.INFO):
            # the report sorts + str()ifies the WHOLE history (reference
            # semantics, timeout_manager.py:63-70) — skip the O(N log N)
            # formatting when nothing consumes it (bench runs)
            return
        elapsed = sorted((t, w, i) for (t, w, i) in self._compute_times
                         if i > self.cdf_start_tracking)
        starts = [t for (s, t) in self._iter_start_times
                  if s > self.cdf_start_tracking]
        iter_times = [starts[i + 1] - starts[i] for i in range(len(starts) - 1)]
        log.info("ELAPSED TIMES %s", str(elapsed))
        log.info("ITERATION TIMES %s", str(iter_times))

    def record_cdf(self, step: int, compute_time_s: float):
        """CDF bookkeeping for steps whose compute ran OUTSIDE reduce()
        (the graph-replayed cdf path at world=1): gather per-rank times,
        append, emit the scraper report — mirrors reduce()'s cdf block."""
        times = self._all_gather_times(compute_time_s, step)
        for w, tm in enumerate(times):
            self._compute_times.append((tm, w, step))
        self._maybe_log_cdf(step)

    def record_cdf_batch(self, steps, my_times):
        """Batched form of record_cdf: one gather for a WINDOW of steps
        (the graph path times every replay with hipEvent pairs but only
        synchronizes once per report window — per-step event.synchronize
        was the cdf mode's bottleneck, ~70 us/step of host round-trips).
        Report content is identical to the per-step form."""
        if not steps:
            return
        if self.distributed:
            t = torch.tensor(my_times, dtype=torch.float64,
                             device=self.flat_grad.device)
            out = torch.zeros(self.world * len(my_times),
                              dtype=torch.float64, device=t.device)
            dist.all_gather_into_tensor(out, t, group=self.group)
            allt = out.view(self.world, -1).tolist()
        else:
            allt = [my_times]
        for w, row in enumerate(allt):
            for s, tm in zip(steps, row):
                self._compute_times.append((tm, w, s))
        self._maybe_log_cdf(steps[-1], force=True)

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
