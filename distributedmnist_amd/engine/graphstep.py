"""hipGraph-captured training step.

The whole step — grad zero, fwd (HIP kernels), bwd, RCCL all-reduce, fused
SGD — is captured once into a hipGraph and replayed per iteration: per-step
host work collapses to two async copies (batch into the static buffers) +
one graph launch.  Host kernel arguments are frozen under replay, so the
pieces that change per step live in DEVICE memory:
  - step_dev (int64): dropout / drop-connect philox offset
  - lr_scale_dev (fp32): staircase LR x 1/contributors
advanced INSIDE the graph by the step_advance kernel at the END of the body
(so the values seen by the body equal the eager schedule's).

Only the full_sync mode is graphable (K-of-N / interval / CDF need
data-dependent host control flow); Trainer falls back to eager elsewhere.

Two capture scopes:
  - FULL (world == 1, or --hip_graph full): the whole step including the
    RCCL all-reduce is one graph.
  - SPLIT (world > 1 default): only the compute body (fwd + two-stream bwd
    -> grads in the flat bucket) is captured; the all-reduce and the
    device-arg optimizer tail (sgd_step_dev, transpose refresh,
    step_advance) run eagerly after each replay.  A collective that a
    backend refuses to capture INVALIDATES the capture and leaves the
    process unrecoverable (see docs/ROUND1_NOTES.md), so the multi-rank
    default never captures one; the cost is ~4 extra launches + an
    un-overlapped collective per step.
"""

from __future__ import annotations

import logging

import torch
import torch.distributed as dist

from .. import _C

log = logging.getLogger("dmnist.graph")


class GraphedStep:
    def __init__(self, trainer, batch_shape, split: bool = False):
        t = trainer
        self.split = split
        assert t.device.type == "cuda", "GraphedStep requires a GPU"
        assert t.engine.timeout_s is None and (
            t.mode == "full_sync" or (t.mode == "cdf" and t.world == 1)), \
            "graph-captured: full_sync, or cdf at world=1 (identical apply " \
            "semantics — contributors == world, no drop path)"
        self.t = t
        ext = _C.ext()
        dev = t.device
        flags = t.flags
        self.static_x = torch.zeros(batch_shape, dtype=t.compute_dtype, device=dev)
        self.static_y = torch.zeros(batch_shape[0], dtype=torch.int64, device=dev)
        self.step_dev = torch.zeros(1, dtype=torch.int64, device=dev)
        self.lr_scale_dev = torch.zeros(1, dtype=torch.float32, device=dev)
        num_batches = t._num_examples / flags.batch_size
        self.decay_steps = max(1, int(num_batches * flags.num_epochs_per_decay
                                      / max(1, t.engine.K)))
        self.inv_contrib = 1.0 / max(1, t.world)
        # pre-aggregation masks run inside the capture (per-rank, keyed on
        # step_dev); only the legacy --drop_connect_post form rides in the
        # SGD tail kernel
        self.dc_pre = t._dc_pre
        self.dc_prob = flags.drop_connect_probability
        self.dc_keep = (flags.drop_connect_probability
                        if (flags.drop_connect and not t._dc_pre) else -1.0)
        self._ext = ext
        # hand-scheduled two-stream step for LeNet (dW off the dX chain);
        # autograd body otherwise
        self._fused = None
        from ..models import LeNet5
        if (isinstance(t.model, LeNet5) and t.model.shadows
                and t.model.shadows_T
                and getattr(flags, "fused_step", "auto") != "off"):
            from .fused_step import FusedLeNetStep
            self._fused = FusedLeNetStep(t)

        # entering from eager steps the bucket may hold stale grads (the
        # eager path zeroes at step START); the graphed protocol zeroes at
        # step END (SGD tail), so establish its invariant once here
        t.fp.flat_grad.zero_()
        # snapshot state: the warmup iterations below really train
        master0 = t.fp.flat_master.clone()
        mom0 = (t.flat_momentum.clone()
                if getattr(t, "flat_momentum", None) is not None else None)
        t.model.set_step_dev(self.step_dev)

        if self.split and self._fused is not None:
            # split scope: the bucketed all-reduce must stay OUT of the
            # captured region (run eagerly after replay instead)
            self._fused.overlap_allreduce = False
        if self.dc_pre and self._fused is not None:
            # pre-aggregation masks must precede the reduce: the fused
            # step's in-body fc all-reduce would fire before the mask, so
            # fall back to the post-body whole-bucket reduce in _body
            # (only reachable via --hip_graph full at world > 1)
            self._fused.overlap_allreduce = False

        self._prime(t.step)
        torch.cuda.synchronize()
        # Warmup is COLLECTIVE-FREE: it trains 3 throwaway steps on zero
        # images whose state is rolled back below, so skipping the
        # all-reduce changes nothing — and it means a rank that failed
        # earlier in __init__ (e.g. static-buffer OOM) cannot leave the
        # surviving ranks hung in a warmup collective.  Divergence after
        # try_graph returns is caught by Trainer.get_graph's agreement
        # all-reduce.
        self._warmup = True
        fused_ar = self._fused.overlap_allreduce if self._fused else False
        if self._fused is not None:
            self._fused.overlap_allreduce = False
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                self._body()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self._warmup = False
        if self._fused is not None:
            self._fused.overlap_allreduce = fused_ar

        # two-graph split: when the hand-scheduled step is available, the
        # capture splits at the fc-grads-ready point so the (eager) fc-slice
        # all-reduce can overlap graph B's conv backward — recovering the
        # overlap that a single split capture serializes.  DMNIST_SPLIT2=0
        # falls back to the single compute graph.
        import os as _os
        self.two_graph = (self.split and self._fused is not None
                          and _os.environ.get("DMNIST_SPLIT2", "1") != "0")
        fp = t.fp
        fc0 = fp.offsets[fp.names.index("fc1_w")]
        self.fc_offset = fc0
        self.fc_slice = fp.flat_grad[fc0:]
        self.conv_slice = fp.flat_grad[:fc0]
        self._comm_stream = torch.cuda.Stream() if self.two_graph else None
        cur_stream = torch.cuda.current_stream()
        self.graph = torch.cuda.CUDAGraph()
        self.graph_b = torch.cuda.CUDAGraph() if self.two_graph else None
        try:
            if self.two_graph:
                with torch.cuda.graph(self.graph):
                    # no grad-bucket fill: the previous replay's SGD tail
                    # zeroed it (zero_grad=True), and warmup left it zeroed
                    loss, acc = self._fused.stage_fc(
                        self.static_x, self.static_y, self.step_dev)
                    if self.dc_pre:
                        # fc grads final at graph A's boundary: mask them
                        # BEFORE the eager fc-slice all-reduce between the
                        # graphs (base = slice offset so the slice-wise
                        # masks equal one whole-buffer mask)
                        ext.grad_mask(self.fc_slice, self.dc_prob,
                                      flags.seed, 0, t.rank,
                                      base=self.fc_offset,
                                      step_dev=self.step_dev)
                    self.static_loss = loss.detach()
                    self.static_acc = acc.detach()  # mean via inv_n
                # graph B shares graph A's memory pool: it reads tensors
                # graph A allocated (the stage_fc stash)
                with torch.cuda.graph(self.graph_b, pool=self.graph.pool()):
                    self._fused.stage_conv()
                    if self.dc_pre:
                        ext.grad_mask(self.conv_slice, self.dc_prob,
                                      flags.seed, 0, t.rank, base=0,
                                      step_dev=self.step_dev)
            else:
                with torch.cuda.graph(self.graph):
                    if self.split:
                        self._body_grads()
                    else:
                        self._body()
        except Exception:
            # torch.cuda.graph.__exit__ calls capture_end() BEFORE popping
            # its stream context: when a body op invalidated the capture,
            # capture_end raises, the thread is left on the dead capture
            # stream, and the capture can stay registered — every later op
            # then fails with hipErrorStreamCaptureUnsupported instead of
            # falling back to eager.  Terminate the capture and restore the
            # stream before propagating (try_graph turns this into eager).
            for g in (self.graph, self.graph_b):
                if g is None:
                    continue
                try:
                    g.capture_end()
                except Exception:
                    pass
            torch.cuda.set_stream(cur_stream)
            # best-effort rollback of the warmup training (3 steps on zero
            # images) so the eager fallback continues from the real weights
            try:
                t.fp.flat_master.copy_(master0)
                if mom0 is not None:
                    t.flat_momentum.copy_(mom0)
                t.fp.sync_shadow()
                t.model.set_step(t.step)
            except Exception:
                pass
            raise

        # restore pre-warmup state
        t.fp.flat_master.copy_(master0)
        if mom0 is not None:
            t.flat_momentum.copy_(mom0)
        t.fp.sync_shadow()
        self._prime(t.step)
        torch.cuda.synchronize()

    def _prime(self, step: int):
        """Set device step + LR for the NEXT body execution."""
        flags = self.t.flags
        self.step_dev.fill_(step)
        lr = (flags.initial_learning_rate *
              flags.learning_rate_decay_factor ** (step // self.decay_steps))
        self.lr_scale_dev.fill_(lr * self.inv_contrib)

    def _body_grads(self):
        """fwd + bwd: gradients land in the flat bucket; returns True if
        the fused step already issued the bucketed all-reduce."""
        t = self.t
        fp = t.fp
        # grad bucket already zero: the SGD tail clears it after consuming
        fused_reduced = False
        if self._fused is not None:
            loss, acc = self._fused(self.static_x, self.static_y,
                                    self.step_dev)
            fused_reduced = self._fused.overlap_allreduce
        else:
            logits = t.model(self.static_x, train=True)
            loss, acc = t.model.loss_and_accuracy(logits, self.static_y)
            loss.backward()
        if self.dc_pre:
            # grads complete, reduce not yet issued: per-rank mask here
            self._ext.grad_mask(fp.flat_grad, self.dc_prob, t.flags.seed,
                                0, t.rank, base=0, step_dev=self.step_dev)
        self.static_loss = loss.detach()
        self.static_acc = acc.detach()
        return fused_reduced

    def _tail(self):
        """optimizer + next-step device args (all args device-resident, so
        this is capturable AND replay-equivalent when run eagerly)."""
        t = self.t
        fp = t.fp
        # zero_grad=True: the bucket is cleared in the same pass that
        # consumes it — no per-step fill kernel in the replayed graph
        self._ext.sgd_step_dev(fp.flat_master, fp.flat_grad,
                               fp.flat_shadow if fp.flat_shadow is not None
                               else fp.flat_master,
                               fp.flat_shadow is not None,
                               self.lr_scale_dev, self.dc_keep,
                               t.flags.seed, self.step_dev,
                               momentum=t.flat_momentum,
                               mu=t.flags.momentum, zero_grad=True)
        if fp._t_pairs:
            # transposes + step/LR advance in ONE dispatch
            self._ext.transpose_bf16_batch_adv(
                [v2 for v2, _ in fp._t_pairs],
                [tt for _, tt in fp._t_pairs],
                self.step_dev, self.lr_scale_dev,
                t.flags.initial_learning_rate,
                t.flags.learning_rate_decay_factor,
                self.decay_steps, self.inv_contrib)
        else:
            fp.refresh_transposes()
            # advance step + LR on-device for the next body execution
            self._ext.step_advance(self.step_dev, self.lr_scale_dev,
                                   t.flags.initial_learning_rate,
                                   t.flags.learning_rate_decay_factor,
                                   self.decay_steps, self.inv_contrib)

    def _body(self):
        t = self.t
        fused_reduced = self._body_grads()
        if (not fused_reduced and not getattr(self, "_warmup", False)
                and t.world > 1 and dist.is_initialized()):
            dist.all_reduce(t.fp.flat_grad, op=dist.ReduceOp.SUM)
        self._tail()

    def run(self, images, labels):
        """Replay one step. Returns (loss, acc) static device tensors."""
        self.static_x.copy_(images, non_blocking=True)
        self.static_y.copy_(labels, non_blocking=True)
        self.graph.replay()
        if self.two_graph:
            t = self.t
            main = torch.cuda.current_stream()
            reduced = t.world > 1 and dist.is_initialized()
            if reduced:
                # fc slice reduces on a comm stream WHILE graph B replays
                # the conv backward on the main stream
                self._comm_stream.wait_stream(main)
                with torch.cuda.stream(self._comm_stream):
                    t.engine.wire_allreduce(self.fc_slice,
                                            start=self.fc_offset)
            self.graph_b.replay()
            if reduced:
                main.wait_stream(self._comm_stream)
                t.engine.wire_allreduce(self.conv_slice, start=0)
            self._tail()
        elif self.split:
            t = self.t
            if t.world > 1 and dist.is_initialized():
                t.engine.wire_allreduce(t.fp.flat_grad)
            self._tail()
        self.t.step += 1
        return self.static_loss, self.static_acc


def try_graph(trainer, batch_shape, split: bool = False):
    """Build a GraphedStep, or None if capture is unsupported here."""
    try:
        return GraphedStep(trainer, batch_shape, split=split)
    except Exception as e:  # noqa: BLE001 — fall back to eager on any failure
        log.warning("hipGraph capture unavailable (%s); running eager", e)
        return None
