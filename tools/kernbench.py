#!/usr/bin/env python3
"""Per-op microbenchmarks of the HIP kernel library (run on a GPU box).

Times each extension op solo with hipEvents (median of N), at the two
training shapes (per-GPU batch 1024 and 8192).  Output is markdown for
profiles/.

Usage: python tools/kernbench.py [batch ...]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from distributedmnist_amd import _C  # noqa: E402

bf16 = torch.bfloat16


def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    times = []
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    for _ in range(iters):
        e0.record()
        fn()
        e1.record()
        e1.synchronize()
        times.append(e0.elapsed_time(e1) * 1000)  # us
    times.sort()
    return times[len(times) // 2]


def bench_batch(ext, B):
    dev = "cuda"
    g = torch.Generator(device="cpu")
    g.manual_seed(0)
    x1 = (torch.rand((B, 28, 28, 1), generator=g) - 0.5).to(dev, bf16)
    w1 = (torch.randn((5, 5, 1, 32), generator=g) * 0.1).to(dev, bf16)
    b1 = torch.randn(32, generator=g).to(dev).float() * 0.1
    w2 = (torch.randn((5, 5, 32, 64), generator=g) * 0.1).to(dev, bf16)
    b2 = torch.randn(64, generator=g).to(dev).float() * 0.1
    w2T = w2.reshape(800, 64).t().contiguous()
    f1 = (torch.randn((3136, 512), generator=g) * 0.05).to(dev, bf16)
    f1T = f1.t().contiguous()
    fb1 = torch.full((512,), 0.1, device=dev)
    f2 = (torch.randn((512, 10), generator=g) * 0.05).to(dev, bf16)
    f2T = f2.t().contiguous()
    fb2 = torch.full((10,), 0.1, device=dev)
    labels = torch.randint(0, 10, (B,), device=dev)

    y1, am1 = ext.conv_pool_fwd(x1, w1, b1, None)
    y2, am2 = ext.conv_pool_fwd(y1, w2, b2, w2T)
    h2 = y2.view(B, 3136)
    a1 = ext.linear_act_fwd(h2, f1, fb1, True, 0.5, 1, 2, wT=f1T)
    logits = ext.linear_act_fwd(a1, f2, fb2, False, 1.0, 0, 0, wT=f2T)
    _, _, dl = ext.softmax_xent_fwd(logits, labels)
    db2f = torch.zeros(10, device=dev)
    dyeff2 = ext.mask_db(dl, dl, False, 1.0, db2f)
    dw2f = torch.zeros(512, 10, device=dev)
    dx2 = ext.linear_dx(dyeff2, f2)
    db1f = torch.zeros(512, device=dev)
    dyeff1 = ext.mask_db(dx2, a1, True, 0.5, db1f)
    dw1f = torch.zeros(3136, 512, device=dev)
    dx1 = ext.linear_dx(dyeff1, f1).view(B, 7, 7, 64)
    dbc2 = torch.zeros(64, device=dev)
    dact2 = ext.pool_scatter(dx1, y2, am2, dbc2, 14, 14)
    dwc2 = torch.zeros(5, 5, 32, 64, device=dev)
    dxc = ext.conv_dx(dact2, w2, 32)
    dbc1 = torch.zeros(32, device=dev)
    dact1 = ext.pool_scatter(dxc, y1, am1, dbc1, 28, 28)
    dwc1 = torch.zeros(5, 5, 1, 32, device=dev)
    master = torch.randn(1663370, device=dev)
    grad = torch.randn(1663370, device=dev)
    shadow = torch.zeros(1663370, dtype=bf16, device=dev)

    rows = [
        ("conv1 fwd (direct VALU slab)", lambda: ext.conv_pool_fwd(x1, w1, b1, None)),
        ("conv2 fwd (MFMA image slab)", lambda: ext.conv_pool_fwd(y1, w2, b2, w2T)),
        ("fc1 fwd+relu+dropout (MFMA, wT)", lambda: ext.linear_act_fwd(h2, f1, fb1, True, 0.5, 1, 2, wT=f1T)),
        ("fc2 fwd (MFMA, wT)", lambda: ext.linear_act_fwd(a1, f2, fb2, False, 1.0, 0, 0, wT=f2T)),
        ("softmax-CE fused fwd+grad", lambda: ext.softmax_xent_fwd(logits, labels)),
        ("softmax-CE +fc2 db fused", lambda: ext.softmax_xent_fwd(logits, labels, db_out=db2f)),
        ("mask+db (fc2)", lambda: ext.mask_db(dl, dl, False, 1.0, db2f)),
        ("fc2 dW", lambda: ext.linear_dw_into(a1, dyeff2, dw2f)),
        ("fc2 dX", lambda: ext.linear_dx(dyeff2, f2)),
        ("fc2 dX+mask+db fused", lambda: ext.linear_dx_mask(dl, f2, a1, db1f, 0.5)),
        ("mask+db (fc1)", lambda: ext.mask_db(dx2, a1, True, 0.5, db1f)),
        ("fc1 dW (A_T split-K)", lambda: ext.linear_dw_into(h2, dyeff1, dw1f)),
        ("fc1 dX", lambda: ext.linear_dx(dyeff1, f1)),
        ("pool2 bwd scatter", lambda: ext.pool_scatter(dx1, y2, am2, dbc2, 14, 14)),
        ("fc1 dX+unpool fused", lambda: ext.linear_dx_unpool(dyeff1, f1, am2, dbc2, 7, 7, 64)),
        ("conv2 dW (slab/gemm tiered)", lambda: ext.conv_dw_into(y1, dact2, dwc2)),
        ("conv2 dX (image slab)", lambda: ext.conv_dx(dact2, w2, 32)),
        ("pool1 bwd scatter", lambda: ext.pool_scatter(dxc, y1, am1, dbc1, 28, 28)),
        ("conv1 dW (slab)", lambda: ext.conv_dw_into(x1, dact1, dwc1)),
        ("conv1 dW+db pooled fused", lambda: ext.conv1_dw_pooled(x1, dxc, am1, dwc1.view(-1), dbc1)),
        ("fused SGD apply (1.66M params)", lambda: ext.sgd_step(master, grad, shadow, True, 0.01, 1.0, -1.0, 0, 0)),
    ]
    print(f"\n## B = {B} (solo, median of 30, us)\n")
    print("| op | us |")
    print("|---|---|")
    total = 0.0
    for name, fn in rows:
        us = timeit(fn)
        total += us
        print(f"| {name} | {us:.1f} |")
    print(f"| **sum (no overlap)** | **{total:.1f}** |")


def main():
    ext = _C.ext()
    batches = [int(a) for a in sys.argv[1:]] or [1024, 8192]
    print("# Kernel microbenchmarks (1x MI355X, bf16, solo per-op timings)")
    for B in batches:
        bench_batch(ext, B)


if __name__ == "__main__":
    main()
