"""Flat parameter/gradient buffers — the one-bucket all-reduce layout.

The whole model is 1,663,370 params (~6.65 MB fp32); fc1_w is 96.5% of it
(SURVEY.md section 2.4).  Gradient aggregation is therefore ONE fused flat
buffer and ONE RCCL all-reduce per step (replacing the reference's 8
per-variable ConditionalAccumulator pushes,
sync_replicas_optimizer_modified.py:342-357) — at this size the collective is
latency-bound, so fewer/larger messages win on xGMI.

FlatParams:
  - flat_master (fp32): parameters re-homed as views of one contiguous buffer
  - flat_grad   (fp32): .grad views of one contiguous buffer (autograd
    accumulates in-place into them)
  - flat_shadow (bf16, GPU): compute copies used by the MFMA kernels,
    refreshed by the fused SGD kernel in the same pass as the update
"""

from __future__ import annotations

import torch


class FlatParams:
    def __init__(self, model, device=None, compute_dtype: torch.dtype = torch.float32):
        names = list(model.PARAM_ORDER)
        params = [getattr(model, n) for n in names]
        if device is None:
            device = params[0].device
        device = torch.device(device)
        self.names = names
        self.numels = [p.numel() for p in params]
        self.shapes = [tuple(p.shape) for p in params]
        total = sum(self.numels)
        self.total = total
        self.flat_master = torch.empty(total, dtype=torch.float32, device=device)
        self.flat_grad = torch.zeros(total, dtype=torch.float32, device=device)
        off = 0
        self.offsets = []
        for p, n in zip(params, self.numels):
            self.offsets.append(off)
            self.flat_master[off:off + n].copy_(p.detach().reshape(-1).float())
            p.data = self.flat_master[off:off + n].view(p.shape)
            p.grad = self.flat_grad[off:off + n].view(p.shape)
            off += n
        self.params = params
        self.compute_dtype = compute_dtype
        self.flat_shadow = None
        if compute_dtype != torch.float32:
            self.flat_shadow = torch.empty(total, dtype=compute_dtype, device=device)
            self.flat_shadow.copy_(self.flat_master.to(compute_dtype))
            model.shadows = {
                n: self.flat_shadow[o:o + sz].view(shape)
                for n, o, sz, shape in zip(names, self.offsets, self.numels, self.shapes)
            }
            # pre-transposed copies of the forward-GEMM B operands: vector
            # LDS staging instead of scatter transposes every K-step
            self.shadow_T = {}
            self._t_pairs = []
            for n in getattr(model, "TRANSPOSED_WEIGHTS", []):
                v = model.shadows[n]
                v2 = v.reshape(-1, v.shape[-1])        # [K][N] k-major
                t = torch.empty(v2.shape[1], v2.shape[0],
                                dtype=compute_dtype, device=device)
                self.shadow_T[n] = t
                self._t_pairs.append((v2, t))
            model.shadows_T = self.shadow_T
            self.refresh_transposes()
        else:
            model.shadows = {}
            model.shadows_T = {}
            self.shadow_T = {}
            self._t_pairs = []

    def zero_grad(self):
        self.flat_grad.zero_()

    def fix_grad_views(self):
        """Autograd normally accumulates in-place into the .grad views; if a
        torch version replaced .grad out-of-place, copy back and re-alias.
        Returns True if all views were intact (no copies needed)."""
        intact = True
        for p, off, n in zip(self.params, self.offsets, self.numels):
            g = p.grad
            view = self.flat_grad[off:off + n].view(p.shape)
            if g is None:
                intact = False
                p.grad = view
            elif g.data_ptr() != view.data_ptr():
                intact = False
                view.copy_(g.detach().float())
                p.grad = view
        return intact

    def sync_shadow(self):
        if self.flat_shadow is not None:
            self.flat_shadow.copy_(self.flat_master.to(self.flat_shadow.dtype))
            self.refresh_transposes()

    def refresh_transposes(self):
        if not self._t_pairs:
            return
        if self.flat_shadow.is_cuda:
            from .. import _C
            ext = _C.ext()
            ext.transpose_bf16_batch([v2 for v2, _ in self._t_pairs],
                                     [t for _, t in self._t_pairs])
        else:
            for v2, t in self._t_pairs:
                t.copy_(v2.t())

    def state_dict_params(self):
        return {n: self.flat_master[o:o + sz].view(shape).clone()
                for n, o, sz, shape in zip(self.names, self.offsets,
                                           self.numels, self.shapes)}

    def load_flat(self, flat: torch.Tensor):
        self.flat_master.copy_(flat.to(self.flat_master.device))
        self.sync_shadow()
