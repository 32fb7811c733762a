"""Autograd wiring for the fused primitives, dispatching CPU-ref vs HIP ext.

Dispatch rule: CUDA tensors -> the in-tree HIP extension (fails loudly if the
.so is missing on a GPU box — _C.ext()); CPU tensors -> ops/cpu_ref.py fp32
reference.  Both paths share autograd semantics so the engine/ and models/
code is device-agnostic.

Mixed precision: parameters are fp32 masters; on GPU the model holds bf16
shadow copies used for compute (passed as the non-differentiable `w_comp` /
`b_comp` args).  Weight gradients are always produced in fp32 so they can
accumulate straight into the flat fp32 all-reduce bucket
(parallel/sync.py).
"""

from __future__ import annotations

import torch

from .. import _C
from . import cpu_ref


def _use_hip(t: torch.Tensor) -> bool:
    # the HIP kernels are the bf16 compute path (bf16 operands, fp32
    # epilogues); fp32-on-GPU routes to the torch reference ops (MIOpen /
    # rocBLAS) — a numerics-debug mode, never the bench path
    return t.is_cuda and t.dtype == torch.bfloat16


class ConvPoolFn(torch.autograd.Function):
    """Fused conv5x5-SAME + bias + ReLU + maxpool2x2s2 (mnist.py:107-127).

    GPU direct-grad mode: when dw_out/db_out (fp32 views into the flat
    all-reduce bucket, pre-zeroed each step) are passed, the backward
    kernels accumulate straight into them and None is returned to autograd
    for w/b — no per-param zeros() or AccumulateGrad adds on the hot path.
    """

    @staticmethod
    def forward(ctx, x, w, b, w_comp, b_comp, need_dx: bool,
                dw_out, db_out, w_t):
        if _use_hip(x):
            y, amax = _C.ext().conv_pool_fwd(x, w_comp, b_comp, w_t)
        else:
            y, amax = cpu_ref.conv_pool_fwd(x, w_comp, b_comp)
        ctx.save_for_backward(x, w_comp, y, amax)
        ctx.need_dx = need_dx
        ctx.outs = (dw_out, db_out) if (dw_out is not None and _use_hip(x)) else None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w_comp, y, amax = ctx.saved_tensors
        if _use_hip(dy):
            if ctx.outs is not None:
                dw_out, db_out = ctx.outs
                dx = _C.ext().conv_pool_bwd_into(
                    dy.contiguous(), x, w_comp, y, amax, ctx.need_dx,
                    dw_out, db_out)
                dw = db = None
            else:
                dx, dw, db = _C.ext().conv_pool_bwd(
                    dy.contiguous(), x, w_comp, y, amax, ctx.need_dx)
        else:
            dx, dw, db = cpu_ref.conv_pool_bwd(dy, x, w_comp, y, amax)
        if not ctx.need_dx:
            dx = None
        return dx, dw, db, None, None, None, None, None, None


def conv_pool(x, w, b, w_comp=None, b_comp=None, need_dx=True,
              dw_out=None, db_out=None, w_t=None):
    return ConvPoolFn.apply(x, w, b, w_comp if w_comp is not None else w,
                            b_comp if b_comp is not None else b, need_dx,
                            dw_out, db_out, w_t)


class LinearActFn(torch.autograd.Function):
    """x @ W + b [+ReLU] [+TF-dropout(1-p_keep)] (mnist.py:136-145).

    Dropout is folded into the epilogue; the backward recovers BOTH the relu
    and the dropout mask from sign(y): dropped or relu-clipped positions have
    y == 0, kept positions are scaled by 1/p_keep.
    """

    @staticmethod
    def forward(ctx, x, w, b, w_comp, b_comp, relu: bool, p_keep: float,
                seed: int, offset: int, dw_out, db_out, offset_dev, w_t):
        if _use_hip(x):
            if offset_dev is not None and p_keep < 1.0:
                y = _C.ext().linear_act_fwd_dev(x, w_comp, b_comp, relu,
                                                p_keep, seed, offset_dev,
                                                wT=w_t)
            else:
                y = _C.ext().linear_act_fwd(x, w_comp, b_comp, relu, p_keep,
                                            seed, offset, wT=w_t)
        else:
            y = cpu_ref.linear_fwd(x, w_comp, b_comp, relu)
            if p_keep < 1.0:
                gen = torch.Generator(device="cpu")
                gen.manual_seed((seed * 0x9E3779B97F4A7C15 + offset) % (2**63))
                mask = (torch.rand(y.shape, generator=gen) < p_keep)
                y = y * mask.to(device=y.device, dtype=y.dtype) / p_keep
        ctx.save_for_backward(x, w_comp, y)
        ctx.relu = relu
        ctx.p_keep = p_keep
        ctx.outs = (dw_out, db_out) if (dw_out is not None and _use_hip(x)) else None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w_comp, y = ctx.saved_tensors
        relu, p_keep = ctx.relu, ctx.p_keep
        if _use_hip(dy):
            if ctx.outs is not None:
                dw_out, db_out = ctx.outs
                dx = _C.ext().linear_act_bwd_into(
                    dy.contiguous(), x, w_comp, y, relu, p_keep,
                    ctx.needs_input_grad[0], dw_out, db_out)
                dw = db = None
            else:
                dx, dw, db = _C.ext().linear_act_bwd(
                    dy.contiguous(), x, w_comp, y, relu, p_keep,
                    ctx.needs_input_grad[0])
        else:
            dyf = dy.float()
            if relu or p_keep < 1.0:
                dyf = dyf * (y > 0).float()
                if p_keep < 1.0:
                    dyf = dyf / p_keep
            dx = (dyf @ w_comp.float().t()).to(x.dtype)
            dw = x.float().t() @ dyf
            db = dyf.sum(dim=0)
        if not ctx.needs_input_grad[0]:
            dx = None
        return (dx, dw, db, None, None, None, None, None, None, None, None,
                None, None)


def linear_act(x, w, b, w_comp=None, b_comp=None, relu=False, p_keep=1.0,
               seed=0, offset=0, dw_out=None, db_out=None, offset_dev=None,
               w_t=None):
    return LinearActFn.apply(x, w, b, w_comp if w_comp is not None else w,
                             b_comp if b_comp is not None else b, relu,
                             float(p_keep), int(seed), int(offset),
                             dw_out, db_out, offset_dev, w_t)


class SoftmaxXentFn(torch.autograd.Function):
    """Fused mean sparse-softmax-CE + top-1-correct count (mnist.py:149-164).

    Forward also produces dlogits = (softmax - onehot)/B so backward is a
    saved-tensor multiply — one fused GPU kernel total.
    Returns (loss, correct_count); correct_count is non-differentiable.
    """

    @staticmethod
    def forward(ctx, logits, labels):
        if _use_hip(logits):
            loss, correct, dl = _C.ext().softmax_xent_fwd(logits, labels)
        else:
            loss, correct, dl = cpu_ref.softmax_xent_fwd(logits, labels)
        ctx.save_for_backward(dl)
        ctx.mark_non_differentiable(correct)
        return loss, correct

    @staticmethod
    def backward(ctx, dloss, _dcorrect):
        (dl,) = ctx.saved_tensors
        return (dl * dloss).to(dl.dtype), None


def softmax_xent(logits, labels):
    return SoftmaxXentFn.apply(logits, labels)


def grad_mask(grad: torch.Tensor, keep: float, seed: int, step: int,
              rank: int, step_dev=None, base: int = 0):
    """Per-rank PRE-AGGREGATION drop-connect: in-place Bernoulli(keep) mask
    of this rank's local gradient BEFORE the all-reduce, seeded by
    (seed, step, rank) so every worker's mask is distinct — the reference's
    estimator (distributed_train.py:194-203: masks applied per worker, then
    averaged).  Mask, NO rescale (:414-416)."""
    if grad.is_cuda:
        _C.ext().grad_mask(grad, float(keep), int(seed), int(step),
                           int(rank), base=int(base), step_dev=step_dev)
    else:
        gen = torch.Generator(device="cpu")
        gen.manual_seed(((seed ^ 0x9D5AD0C5) * 0x9E3779B97F4A7C15
                         + (step << 20) + rank + 1) % (2**63))
        mask = (torch.rand(grad.shape, generator=gen) < keep).float()
        grad.mul_(mask.to(grad.device))


def sgd_step(master: torch.Tensor, grad: torch.Tensor, lr: float,
             grad_scale: float = 1.0, drop_connect_keep=None,
             seed: int = 0, offset: int = 0, shadow=None,
             momentum=None, mu: float = 0.0):
    """Fused flat SGD apply: master -= lr*scale*(grad [* bernoulli(keep)]).

    Optionally refreshes the bf16 `shadow` copy (one kernel on GPU).
    Reference semantics: GradientDescentOptimizer + drop_connect mask with NO
    rescale (distributed_train.py:176,414-416).
    """
    if master.is_cuda:
        _C.ext().sgd_step(master, grad,
                          shadow if shadow is not None else master,
                          shadow is not None, float(lr), float(grad_scale),
                          -1.0 if drop_connect_keep is None else float(drop_connect_keep),
                          int(seed), int(offset),
                          momentum=momentum, mu=float(mu))
    else:
        gen = None
        if drop_connect_keep is not None:
            gen = torch.Generator(device="cpu")
            gen.manual_seed((seed * 0x9E3779B97F4A7C15 + offset) % (2**63))
        cpu_ref.sgd_step(master, grad, lr, grad_scale,
                         drop_connect_keep, gen, shadow,
                         momentum=momentum, mu=mu)
