"""Hand-scheduled LeNet training step (GPU bf16 path).

Replaces autograd's serial backward with an explicit schedule; used inside
the hipGraph capture (graphstep.py).

Backward dataflow (round 2, pool backward fused away):

  softmax(+db2) -> dl --> dW2
     dl --> dx2+mask+db1 = dyeff1 --> dW1
              dyeff1 --> dX+unpool = dact2 (+db_conv2) --> conv2 dW
                          dact2 --> conv_dx = dxc --> conv1 dW+db (pooled)
All dW/db land directly in the flat fp32 all-reduce bucket (zeroed by the
SGD tail of the previous step).

Round-1 ran the dW chain on a SIDE stream overlapping the dX chain;
round-2 same-box A/B showed the single-stream schedule is ~7% faster
end-to-end at B<=2048 (cross-stream graph-edge semaphores + CU contention
on the dX chain outweigh the overlap there) while the overlap still wins at
B>=4096 — the schedule is batch-tiered (FusedLeNetStep.__init__).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import _C


class FusedLeNetStep:
    def __init__(self, trainer):
        t = trainer
        m = t.model
        from ..models import LeNet5
        assert isinstance(m, LeNet5), "fused step is LeNet-specific"
        assert t.compute_dtype == torch.bfloat16 and t.device.type == "cuda"
        assert m.shadows and m.shadows_T, "fused step needs bf16 shadows"
        self.t = t
        self.ext = _C.ext()
        # Stream schedule is BATCH-TIERED (same-box A/B, after the
        # per-call side-stream fix): at B=1024 the cross-stream fork/join
        # edges of the replayed graph plus the dW kernels' CU contention on
        # the dX chain cost more than the overlap buys (0.429 two-stream vs
        # 0.400 ms single; 0.384 after the conv1-dW G retune); at B=8192
        # the hidden dW work dominates the edge cost (1.698 vs 1.733) —
        # single-stream below 4096, overlapped above.  DMNIST_TWO_STREAM /
        # DMNIST_SINGLE_STREAM force either for A/B.
        # overlap level: 0 = single stream, 1 = fc dW only on the side
        # stream (cheap contention, conv2 dW serial), 2 = every dW on the
        # side stream (round-1 schedule).  DMNIST_OVERLAP=0/1/2 forces;
        # legacy DMNIST_TWO_STREAM=1 -> 2, DMNIST_SINGLE_STREAM=1 -> 0.
        import os as _os
        if _os.environ.get("DMNIST_OVERLAP"):
            self.overlap_level = int(_os.environ["DMNIST_OVERLAP"])
        elif _os.environ.get("DMNIST_TWO_STREAM"):
            self.overlap_level = 2
        elif _os.environ.get("DMNIST_SINGLE_STREAM"):
            self.overlap_level = 0
        else:
            self.overlap_level = 0 if t.flags.batch_size < 4096 else 2
        self.single_stream = self.overlap_level == 0
        # side stream resolved PER CALL: in single-stream mode it must be
        # whatever stream the body is running on RIGHT NOW (warmup and
        # graph capture run on their own streams — freezing the init-time
        # stream here sent the dW ops to a foreign stream with no
        # dependency edges: silent gradient races)
        self.side = None if self.single_stream else torch.cuda.Stream()
        self.p_keep = 0.5
        self.seed = t.flags.seed
        # bucketed all-reduce: the fc gradients (96.5% of the payload,
        # SURVEY.md §2.4) are complete long before the conv backward — their
        # collective overlaps it on the side stream.  The flat bucket layout
        # (conv*, then fc*) makes the split two contiguous slices.
        fp = t.fp
        fc0 = fp.offsets[fp.names.index("fc1_w")]
        self.fc_offset = fc0
        self.conv_slice = fp.flat_grad[:fc0]
        self.fc_slice = fp.flat_grad[fc0:]
        self.overlap_allreduce = t.world > 1 and dist.is_initialized()

    def _fork(self, a, b):
        """a.wait_stream(b), elided in single-stream mode (each edge is a
        semaphore pair in the captured graph)."""
        if not self.single_stream:
            a.wait_stream(b)

    def __call__(self, x, labels, step_dev):
        """One fwd+bwd; gradients land in the flat bucket.
        Returns (loss, mean_accuracy) device scalars."""
        ext = self.ext
        m = self.t.model
        sh, shT = m.shadows, m.shadows_T
        B = x.shape[0]

        def gv(name):
            return getattr(m, name).grad

        s0 = torch.cuda.current_stream()
        s1 = s0 if self.single_stream else self.side

        # ---- forward (s0) ----
        y1, am1 = ext.conv_pool_fwd(x, sh["conv1_w"], m.conv1_b, None)
        y2, am2 = ext.conv_pool_fwd(y1, sh["conv2_w"], m.conv2_b,
                                    shT["conv2_w"])
        h2 = y2.view(B, 7 * 7 * 64)
        a1 = ext.linear_act_fwd_dev(h2, sh["fc1_w"], m.fc1_b, True,
                                    self.p_keep, self.seed, step_dev,
                                    shT["fc1_w"])
        logits = ext.linear_act_fwd(a1, sh["fc2_w"], m.fc2_b, False, 1.0,
                                    0, 0, shT["fc2_w"])
        # softmax computes the fc2 bias grad in the same pass (column sums
        # of dlogits) — no standalone mask_db launch for fc2
        # inv_n folds the accuracy mean into the kernel: no scalar-divide
        # launch per step; `acc` below is already the MEAN accuracy
        loss, acc, dl = ext.softmax_xent_fwd(logits, labels,
                                             db_out=gv("fc2_b"),
                                             inv_n=1.0 / B)

        # ---- backward: dX chain on s0, dW GEMMs on s1 ----
        self._fork(s1, s0)
        with torch.cuda.stream(s1):
            ext.linear_dw_into(a1, dl, gv("fc2_w"))
        # fc2 dX with the fc1 relu+dropout mask folded into the epilogue
        # (+ fc1 bias grad) — replaces linear_dx + mask_db
        dyeff1 = ext.linear_dx_mask(dl, sh["fc2_w"], a1, gv("fc1_b"),
                                    self.p_keep)
        self._fork(s1, s0)
        with torch.cuda.stream(s1):
            ext.linear_dw_into(h2, dyeff1, gv("fc1_w"))
            if self.overlap_allreduce:
                # every fc gradient (fc2 via the earlier s1 work + the db's
                # ordered by wait_stream) is final here
                self.t.engine.wire_allreduce(self.fc_slice,
                                             start=self.fc_offset)
        # fc1 dX fused with pool2 backward: dact2 + conv2 db come straight
        # out of the GEMM epilogue (no pool_scatter kernel on the chain;
        # liveness rides in the amax byte, no y2 re-read)
        dact2 = ext.linear_dx_unpool(dyeff1, sh["fc1_w"], am2,
                                     gv("conv2_b"), 7, 7, 64)
        sW = s1 if self.overlap_level >= 2 else s0  # conv2 dW placement
        if sW is not s0:
            self._fork(sW, s0)
        with torch.cuda.stream(sW):
            ext.conv_dw_into(y1, dact2, gv("conv2_w"))
        dxc = ext.conv_dx(dact2, sh["conv2_w"], 32)

        # conv1 dW+db consume the POOLED gradient (pool1 backward fused in
        # the consumer; the 4x-size dact1 is never materialized)
        ext.conv1_dw_pooled(x, dxc, am1, gv("conv1_w"), gv("conv1_b"))

        self._fork(s0, s1)
        if self.overlap_allreduce:
            self.t.engine.wire_allreduce(self.conv_slice, start=0)
        # keep the side-stream consumers alive until the join (capture-safe)
        self._keep = (a1, h2, dyeff1, dl, y1, dact2)
        return loss, acc

    # ------------------------------------------------------------------
    # Two-stage form of the SAME schedule, for the two-graph split capture
    # (graphstep.py): stage_fc ends with every fc gradient final (so the
    # eager fc-slice all-reduce can launch between the graphs and overlap
    # stage_conv's replay); stage_conv finishes the conv backward.  The
    # math and stream schedule per stage are identical to __call__ above —
    # which stays the single-capture/eager body — except the collectives
    # live OUTSIDE and the fc stage joins its side stream at the end so
    # graph A's boundary is well-defined.
    def stage_fc(self, x, labels, step_dev):
        ext = self.ext
        m = self.t.model
        sh, shT = m.shadows, m.shadows_T
        B = x.shape[0]

        def gv(name):
            return getattr(m, name).grad

        s0 = torch.cuda.current_stream()
        s1 = s0 if self.single_stream else self.side
        y1, am1 = ext.conv_pool_fwd(x, sh["conv1_w"], m.conv1_b, None)
        y2, am2 = ext.conv_pool_fwd(y1, sh["conv2_w"], m.conv2_b,
                                    shT["conv2_w"])
        h2 = y2.view(B, 7 * 7 * 64)
        a1 = ext.linear_act_fwd_dev(h2, sh["fc1_w"], m.fc1_b, True,
                                    self.p_keep, self.seed, step_dev,
                                    shT["fc1_w"])
        logits = ext.linear_act_fwd(a1, sh["fc2_w"], m.fc2_b, False, 1.0,
                                    0, 0, shT["fc2_w"])
        # inv_n folds the accuracy mean into the kernel: no scalar-divide
        # launch per step; `acc` below is already the MEAN accuracy
        loss, acc, dl = ext.softmax_xent_fwd(logits, labels,
                                             db_out=gv("fc2_b"),
                                             inv_n=1.0 / B)
        self._fork(s1, s0)
        with torch.cuda.stream(s1):
            ext.linear_dw_into(a1, dl, gv("fc2_w"))
        dyeff1 = ext.linear_dx_mask(dl, sh["fc2_w"], a1, gv("fc1_b"),
                                    self.p_keep)
        self._fork(s1, s0)
        with torch.cuda.stream(s1):
            ext.linear_dw_into(h2, dyeff1, gv("fc1_w"))
        # fused dX+pool2-backward: conv2_b lands here (conv slice — reduced
        # after graph B, so computing it early is safe)
        dact2 = ext.linear_dx_unpool(dyeff1, sh["fc1_w"], am2,
                                     gv("conv2_b"), 7, 7, 64)
        s0.wait_stream(s1)  # graph A boundary: fc grads complete
        self._stash = (x, y1, am1, dact2)
        self._keep = (a1, h2, dyeff1, dl)
        return loss, acc

    def stage_conv(self):
        ext = self.ext
        m = self.t.model

        def gv(name):
            return getattr(m, name).grad

        x, y1, am1, dact2 = self._stash
        s0 = torch.cuda.current_stream()
        s1 = s0 if self.single_stream else self.side
        sW = s1 if self.overlap_level >= 2 else s0  # conv2 dW placement
        if sW is not s0:
            self._fork(sW, s0)
        with torch.cuda.stream(sW):
            ext.conv_dw_into(y1, dact2, gv("conv2_w"))
        dxc = ext.conv_dx(dact2, m.shadows["conv2_w"], 32)
        ext.conv1_dw_pooled(x, dxc, am1, gv("conv1_w"), gv("conv1_b"))
        self._fork(s0, s1)
        self._keep2 = (dact2,)
