"""fp32 PyTorch reference implementations of every hot-path primitive.

These are (a) the CPU execution path (no GPU required — BASELINE config 1),
and (b) the numerics oracle the HIP/CDNA4 kernels are unit-tested against
(SURVEY.md section 4).  Layouts follow the reference model (TF NHWC / HWIO,
/root/reference/src/mnist.py:107-145): activations are NHWC, conv weights are
[KH, KW, Cin, Cout], fc weights are [in, out] with y = x @ W + b.

All functions are pure tensor->tensor (no autograd); autograd wiring lives in
ops/functional.py.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# conv 5x5 SAME + bias + ReLU + maxpool 2x2 s2, fused (fwd + bwd)
# Reference call sites: mnist.py:107-127 (conv/bias_add/relu/max_pool x2).
# ---------------------------------------------------------------------------

def conv_pool_fwd(x, w, b):
    """x: [N,H,W,Cin], w: [5,5,Cin,Cout], b: [Cout] ->
    (y: [N,H/2,W/2,Cout], argmax: [N,H/2,W/2,Cout] uint8 in 0..3)

    y = maxpool2x2(relu(conv2d_same(x, w) + b)); argmax records which of the
    2x2 window (r*2+c) won, for routing the pooled gradient back.
    """
    N, H, W, Cin = x.shape
    Cout = w.shape[3]
    xc = x.permute(0, 3, 1, 2)                       # NCHW
    wc = w.permute(3, 2, 0, 1)                       # [Cout,Cin,5,5]
    conv = F.conv2d(xc.float(), wc.float(), b.float(), padding=2)
    act = F.relu(conv)                               # [N,Cout,H,W]
    # maxpool 2x2 stride 2 with argmax
    pooled, idx = F.max_pool2d(act, 2, 2, return_indices=True)
    # idx is flat index into H*W; convert to 0..3 position within the window
    Ho, Wo = pooled.shape[2], pooled.shape[3]
    ar = torch.arange(Ho, device=x.device).view(1, 1, Ho, 1)
    ac = torch.arange(Wo, device=x.device).view(1, 1, 1, Wo)
    r = idx // W - ar * 2
    c = idx % W - ac * 2
    amax = (r * 2 + c).to(torch.uint8)               # [N,Cout,Ho,Wo]
    # liveness rides in the argmax byte (matches the HIP kernels): 7 marks
    # a dead window (relu zeroed all four) so pooled-consumer backward
    # kernels need no y read
    amax = torch.where(pooled > 0, amax,
                       torch.tensor(7, dtype=torch.uint8))
    y = pooled.permute(0, 2, 3, 1).contiguous()      # NHWC
    amax = amax.permute(0, 2, 3, 1).contiguous()
    return y, amax


def conv_pool_bwd(dy, x, w, y, amax):
    """Backward of conv_pool_fwd. Returns (dx, dw, db).

    dy, y, amax: [N,Ho,Wo,Cout]; x: [N,H,W,Cin]; w: [5,5,Cin,Cout].
    ReLU mask: the pooled max is the relu output at the argmax position, so
    grad passes iff y > 0.
    """
    N, Ho, Wo, Cout = dy.shape
    H, W = Ho * 2, Wo * 2
    Cin = x.shape[3]
    # route pooled grad (masked by relu) back to the argmax position
    g = (dy * (y > 0)).permute(0, 3, 1, 2).float()   # [N,Cout,Ho,Wo]
    # clamp the dead-window marker (7) to a valid position: its g is zero
    # by the relu mask, so the scatter target is irrelevant but must be
    # in-bounds
    am = amax.permute(0, 3, 1, 2).long().clamp(max=3)
    dact = x.new_zeros((N, Cout, H, W), dtype=torch.float32)
    ar = torch.arange(Ho, device=x.device).view(1, 1, Ho, 1)
    ac = torch.arange(Wo, device=x.device).view(1, 1, 1, Wo)
    rows = ar * 2 + am // 2
    cols = ac * 2 + am % 2
    flat = (rows * W + cols)
    dact.view(N, Cout, H * W).scatter_(2, flat.view(N, Cout, -1),
                                       g.view(N, Cout, -1))
    xc = x.permute(0, 3, 1, 2).float()               # [N,Cin,H,W]
    wc = w.permute(3, 2, 0, 1).float()               # [Cout,Cin,5,5]
    dxc = torch.nn.grad.conv2d_input(xc.shape, wc, dact, padding=2)
    dwc = torch.nn.grad.conv2d_weight(xc, wc.shape, dact, padding=2)
    db = dact.sum(dim=(0, 2, 3))
    dx = dxc.permute(0, 2, 3, 1).contiguous().to(x.dtype)
    dw = dwc.permute(2, 3, 1, 0).contiguous()        # [5,5,Cin,Cout] fp32
    return dx, dw, db


# ---------------------------------------------------------------------------
# Linear (+ bias [+ ReLU] [+ dropout]) — mnist.py:136-145
# ---------------------------------------------------------------------------

def linear_fwd(x, w, b, relu: bool = False):
    """x: [B,K], w: [K,N], b: [N] -> y: [B,N] (= relu(x@w+b) if relu)."""
    y = x.float() @ w.float() + b.float()
    if relu:
        y = F.relu(y)
    return y.to(x.dtype)


def linear_bwd(dy, x, w, y=None, relu: bool = False):
    """Returns (dx, dw fp32, db fp32). If relu, y (post-relu) masks dy."""
    dyf = dy.float()
    if relu:
        dyf = dyf * (y > 0)
    dx = (dyf @ w.float().t()).to(x.dtype)
    dw = x.float().t() @ dyf
    db = dyf.sum(dim=0)
    return dx, dw, db


def dropout_fwd(x, p_drop: float, gen: torch.Generator | None = None):
    """TF-style dropout (mnist.py:139-140): keep with prob 1-p, scale by
    1/(1-p). Returns (y, mask bool)."""
    keep = 1.0 - p_drop
    # generator lives on CPU; sample there and move (device-agnostic ref)
    mask = (torch.rand(x.shape, device="cpu", dtype=torch.float32,
                       generator=gen) < keep).to(x.device)
    y = x * mask.to(x.dtype) / keep
    return y.to(x.dtype), mask


def dropout_bwd(dy, mask, p_drop: float):
    keep = 1.0 - p_drop
    return (dy * mask.to(dy.dtype) / keep).to(dy.dtype)


# ---------------------------------------------------------------------------
# Fused softmax cross-entropy (+ accuracy): mnist.py:149-164
# ---------------------------------------------------------------------------

def softmax_xent_fwd(logits, labels):
    """logits: [B,C], labels: [B] int64.

    Returns (loss_mean fp32 scalar, correct_count fp32 scalar, dlogits
    [B,C] in logits.dtype).  dlogits = (softmax - onehot)/B — precomputed in
    the forward since the backward needs nothing else (single fused kernel on
    GPU; logsumexp trick).
    """
    lf = logits.float()
    m = lf.max(dim=1, keepdim=True).values
    z = lf - m
    ez = z.exp()
    se = ez.sum(dim=1, keepdim=True)
    logp = z - se.log()
    B = logits.shape[0]
    loss = -logp.gather(1, labels.view(-1, 1)).mean()
    p = ez / se
    dl = p
    dl.scatter_add_(1, labels.view(-1, 1), torch.full((B, 1), -1.0, device=logits.device))
    dl = (dl / B).to(logits.dtype)
    correct = (lf.argmax(dim=1) == labels).float().sum()
    return loss, correct, dl


# ---------------------------------------------------------------------------
# Fused SGD apply (+ drop-connect): distributed_train.py:176,414-416
# ---------------------------------------------------------------------------

def sgd_step(master, grad, lr: float, grad_scale: float = 1.0,
             drop_connect_keep: float | None = None,
             gen: torch.Generator | None = None, shadow=None,
             momentum=None, mu: float = 0.0):
    """In-place w -= lr * grad_scale * (g or momentum-filtered g).

    drop_connect_keep: if set, per-element Bernoulli(keep) mask on the grad
    (reference drop_connect: grad * mask, NO rescale —
    distributed_train.py:414-416).
    momentum/mu: optional heavyweight-ball momentum v = mu*v + g; w -= lr*v
    (the reference used plain GradientDescentOptimizer; momentum is the
    north-star's optional extension).
    shadow: optional flat low-precision copy refreshed after the update.
    """
    g = grad.float() * grad_scale
    if drop_connect_keep is not None:
        mask = (torch.rand(g.shape, device=g.device, dtype=torch.float32,
                           generator=gen) < drop_connect_keep).float()
        g = g * mask
    if momentum is not None:
        momentum.mul_(mu).add_(g)
        g = momentum
    master.add_(g, alpha=-lr)
    if shadow is not None:
        shadow.copy_(master.to(shadow.dtype))
    return master
