"""Unit tests of the fp32 reference primitives against independently-built
torch compositions (these same primitives are the oracle for the HIP
kernels, so they must be right)."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from distributedmnist_amd.ops import cpu_ref
from distributedmnist_amd.ops import functional as Fx


def _torch_conv_block(x_nhwc, w_hwio, b, pool=True):
    """Independent NCHW composition: conv SAME + bias + relu + maxpool."""
    x = x_nhwc.permute(0, 3, 1, 2)
    w = w_hwio.permute(3, 2, 0, 1)
    y = F.relu(F.conv2d(x, w, b, padding=2))
    if pool:
        y = F.max_pool2d(y, 2, 2)
    return y.permute(0, 2, 3, 1)


@pytest.mark.parametrize("N,H,W,Cin,Cout", [(4, 28, 28, 1, 32), (3, 14, 14, 32, 64)])
def test_conv_pool_fwd(N, H, W, Cin, Cout):
    torch.manual_seed(0)
    x = torch.randn(N, H, W, Cin)
    w = torch.randn(5, 5, Cin, Cout) * 0.1
    b = torch.randn(Cout) * 0.1
    y, amax = cpu_ref.conv_pool_fwd(x, w, b)
    ref = _torch_conv_block(x, w, b)
    assert y.shape == (N, H // 2, W // 2, Cout)
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-4)
    assert amax.dtype == torch.uint8
    vals = set(torch.unique(amax).tolist())
    assert vals <= {0, 1, 2, 3, 7}, vals  # 7 = dead-window liveness marker


def test_conv_pool_bwd_matches_autograd():
    torch.manual_seed(1)
    N, H, W, Cin, Cout = 3, 14, 14, 8, 16
    x = torch.randn(N, H, W, Cin).requires_grad_(True)
    w = (torch.randn(5, 5, Cin, Cout) * 0.1).requires_grad_(True)
    b = (torch.randn(Cout) * 0.1).requires_grad_(True)
    ref = _torch_conv_block(x, w, b)
    dy = torch.randn_like(ref)
    ref.backward(dy)

    y, amax = cpu_ref.conv_pool_fwd(x.detach(), w.detach(), b.detach())
    dx, dw, db = cpu_ref.conv_pool_bwd(dy, x.detach(), w.detach(), y, amax)
    torch.testing.assert_close(dx, x.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dw, w.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(db, b.grad, rtol=1e-4, atol=1e-4)


def test_linear_act_autograd():
    torch.manual_seed(2)
    B, K, N = 16, 64, 32
    x = torch.randn(B, K).requires_grad_(True)
    w = (torch.randn(K, N) * 0.1).requires_grad_(True)
    b = (torch.randn(N) * 0.1).requires_grad_(True)
    y = Fx.linear_act(x, w, b, relu=True)
    ref = F.relu(x @ w + b)
    torch.testing.assert_close(y, ref, rtol=1e-5, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    F.relu(x2 @ w2 + b2).backward(dy)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(w.grad, w2.grad, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(b.grad, b2.grad, rtol=1e-5, atol=1e-5)


def test_linear_act_dropout_semantics():
    torch.manual_seed(3)
    B, K, N = 64, 32, 512
    x = torch.randn(B, K)
    w = torch.randn(K, N) * 0.1
    b = torch.zeros(N)
    y = Fx.linear_act(x, w, b, relu=True, p_keep=0.5, seed=7, offset=1)
    base = F.relu(x @ w + b)
    kept = y != 0
    # TF dropout: kept values scaled by 1/keep
    torch.testing.assert_close(y[kept], base[kept] * 2.0, rtol=1e-4, atol=1e-4)
    frac = kept.float().mean().item()
    # base has ~half zeros from relu already; among positives, ~50% kept
    pos = base > 0
    keep_rate = (y[pos] != 0).float().mean().item()
    assert 0.40 < keep_rate < 0.60, keep_rate
    # deterministic per (seed, offset)
    y2 = Fx.linear_act(x, w, b, relu=True, p_keep=0.5, seed=7, offset=1)
    torch.testing.assert_close(y, y2)
    y3 = Fx.linear_act(x, w, b, relu=True, p_keep=0.5, seed=7, offset=2)
    assert not torch.equal(y, y3)


def test_dropout_backward_scaling():
    torch.manual_seed(4)
    B, K, N = 32, 16, 64
    x = torch.randn(B, K).requires_grad_(True)
    w = (torch.randn(K, N) * 0.1).requires_grad_(True)
    b = torch.zeros(N, requires_grad=True)
    y = Fx.linear_act(x, w, b, relu=True, p_keep=0.5, seed=9, offset=0)
    dy = torch.ones_like(y)
    y.backward(dy)
    # dx through kept units only, scaled 1/keep: check via direct recompute
    base = F.relu(x.detach() @ w.detach() + b.detach())
    mask = (y.detach() != 0) & (base > 0)
    dyeff = dy * mask.float() / 0.5
    dw_ref = x.detach().t() @ dyeff
    torch.testing.assert_close(w.grad, dw_ref, rtol=1e-4, atol=1e-4)


def test_softmax_xent_matches_torch():
    torch.manual_seed(5)
    B, C = 64, 10
    logits = torch.randn(B, C, requires_grad=True)
    labels = torch.randint(0, C, (B,))
    loss, correct = Fx.softmax_xent(logits, labels)
    ref = F.cross_entropy(logits, labels)
    torch.testing.assert_close(loss, ref, rtol=1e-5, atol=1e-6)
    acc_ref = (logits.argmax(1) == labels).float().sum()
    assert float(correct) == float(acc_ref)
    loss.backward()
    l2 = logits.detach().clone().requires_grad_(True)
    F.cross_entropy(l2, labels).backward()
    torch.testing.assert_close(logits.grad, l2.grad, rtol=1e-5, atol=1e-6)


def test_sgd_step_basic():
    torch.manual_seed(6)
    master = torch.randn(1000)
    orig = master.clone()
    grad = torch.randn(1000)
    Fx.sgd_step(master, grad, lr=0.1, grad_scale=0.5)
    torch.testing.assert_close(master, orig - 0.05 * grad)


def test_sgd_step_drop_connect():
    torch.manual_seed(7)
    n = 200000
    master = torch.zeros(n)
    grad = torch.ones(n)
    Fx.sgd_step(master, grad, lr=1.0, drop_connect_keep=0.9, seed=3, offset=5)
    # each element either -1 (kept, NO rescale) or 0 (dropped)
    vals = set(torch.unique(master).tolist())
    assert vals <= {-1.0, 0.0}
    keep_rate = (master != 0).float().mean().item()
    assert 0.88 < keep_rate < 0.92
    # deterministic
    m2 = torch.zeros(n)
    Fx.sgd_step(m2, grad, lr=1.0, drop_connect_keep=0.9, seed=3, offset=5)
    torch.testing.assert_close(master, m2)


def test_sgd_step_shadow_refresh():
    master = torch.randn(64)
    shadow = torch.zeros(64, dtype=torch.bfloat16)
    grad = torch.randn(64)
    Fx.sgd_step(master, grad, lr=0.01, shadow=shadow)
    torch.testing.assert_close(shadow, master.to(torch.bfloat16))


def test_sgd_momentum():
    """v = mu*v + scale*g; w -= lr*v (heavyweight-ball)."""
    master = torch.zeros(16)
    mom = torch.zeros(16)
    g = torch.ones(16)
    Fx.sgd_step(master, g, lr=0.1, grad_scale=1.0, momentum=mom, mu=0.9)
    torch.testing.assert_close(master, torch.full((16,), -0.1))
    torch.testing.assert_close(mom, torch.ones(16))
    Fx.sgd_step(master, g, lr=0.1, grad_scale=1.0, momentum=mom, mu=0.9)
    torch.testing.assert_close(mom, torch.full((16,), 1.9))
    torch.testing.assert_close(master, torch.full((16,), -0.29))


def test_trainer_momentum_flag(tmp_path):
    from distributedmnist_amd.engine.train import Trainer, make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--train_dir", str(tmp_path), "--batch_size",
         "16", "--max_steps", "4", "--model", "mlp", "--device", "cpu",
         "--momentum", "0.9", "--save_interval_secs", "100000"])
    t = Trainer(flags)
    assert t.flat_momentum is not None
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    t.train(ds)
    assert float(t.flat_momentum.abs().sum()) > 0


def test_grad_mask_cpu_deterministic_and_rank_distinct():
    """CPU reference of the pre-aggregation drop-connect mask: keep-rate,
    determinism in (seed, step, rank), distinct streams across ranks."""
    import torch
    from distributedmnist_amd.ops import functional as Fx
    g = torch.ones(100_000)
    Fx.grad_mask(g, 0.9, seed=66478, step=3, rank=0)
    kept = float((g != 0).float().mean())
    assert 0.89 < kept < 0.91
    g2 = torch.ones(100_000)
    Fx.grad_mask(g2, 0.9, seed=66478, step=3, rank=0)
    assert torch.equal(g, g2)
    g3 = torch.ones(100_000)
    Fx.grad_mask(g3, 0.9, seed=66478, step=3, rank=1)
    assert not torch.equal(g, g3)


def test_trainer_drop_connect_pre_is_default(tmp_path):
    from distributedmnist_amd.engine.train import Trainer
    from distributedmnist_amd.utils.flags import build_train_parser
    mk = lambda *extra: build_train_parser().parse_args(
        ["--synthetic_data", "--train_dir", str(tmp_path / "t"),
         "--batch_size", "8", "--model", "mlp", "--device", "cpu",
         "--save_interval_secs", "100000"] + list(extra))
    assert Trainer(mk("--drop_connect"))._dc_pre
    assert not Trainer(mk("--drop_connect", "--drop_connect_post"))._dc_pre
    assert not Trainer(mk())._dc_pre
