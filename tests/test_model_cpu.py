"""LeNet/MLP model tests: parity with an independently-built torch module
(same weights) and end-to-end gradient agreement."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from distributedmnist_amd.models import LeNet5, MLP, build_model


class TorchLeNet(nn.Module):
    """Independent NCHW construction sharing the LeNet5 weights."""

    def __init__(self, m: LeNet5):
        super().__init__()
        self.c1w = m.conv1_w.detach().permute(3, 2, 0, 1).clone().requires_grad_(True)
        self.c1b = m.conv1_b.detach().clone().requires_grad_(True)
        self.c2w = m.conv2_w.detach().permute(3, 2, 0, 1).clone().requires_grad_(True)
        self.c2b = m.conv2_b.detach().clone().requires_grad_(True)
        self.f1w = m.fc1_w.detach().clone().requires_grad_(True)
        self.f1b = m.fc1_b.detach().clone().requires_grad_(True)
        self.f2w = m.fc2_w.detach().clone().requires_grad_(True)
        self.f2b = m.fc2_b.detach().clone().requires_grad_(True)

    def forward(self, x_nhwc):
        x = x_nhwc.permute(0, 3, 1, 2)
        h = F.max_pool2d(F.relu(F.conv2d(x, self.c1w, self.c1b, padding=2)), 2, 2)
        h = F.max_pool2d(F.relu(F.conv2d(h, self.c2w, self.c2b, padding=2)), 2, 2)
        h = h.permute(0, 2, 3, 1).reshape(x.shape[0], -1)  # NHWC flatten order
        h = F.relu(h @ self.f1w + self.f1b)
        return h @ self.f2w + self.f2b


def test_lenet_forward_matches_torch():
    torch.manual_seed(0)
    m = LeNet5()
    tm = TorchLeNet(m)
    x = torch.rand(8, 28, 28, 1) - 0.5
    y = m(x, train=False)          # train=False => no dropout
    y_ref = tm(x)
    torch.testing.assert_close(y, y_ref, rtol=1e-4, atol=1e-4)


def test_lenet_grads_match_torch():
    torch.manual_seed(1)
    m = LeNet5()
    tm = TorchLeNet(m)
    x = torch.rand(8, 28, 28, 1) - 0.5
    labels = torch.randint(0, 10, (8,))
    logits = m(x, train=False)
    loss, _ = m.loss_and_accuracy(logits, labels)
    loss.backward()
    ref_logits = tm(x)
    ref_loss = F.cross_entropy(ref_logits, labels)
    ref_loss.backward()
    torch.testing.assert_close(loss.float(), ref_loss, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(m.fc1_w.grad, tm.f1w.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(m.fc2_b.grad, tm.f2b.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(m.conv2_w.grad,
                               tm.c2w.grad.permute(2, 3, 1, 0), rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(m.conv1_b.grad, tm.c1b.grad, rtol=1e-4, atol=1e-5)
    # conv1_w grad exists even though dx is skipped for the first layer
    assert m.conv1_w.grad is not None
    torch.testing.assert_close(m.conv1_w.grad,
                               tm.c1w.grad.permute(2, 3, 1, 0), rtol=1e-4, atol=1e-5)


def test_lenet_init_reference_semantics():
    m = LeNet5()
    assert float(m.conv1_b.abs().sum()) == 0.0
    assert torch.all(m.conv2_b == 0.1)
    assert torch.all(m.fc1_b == 0.1)
    # truncated at 2 sigma
    for w in (m.conv1_w, m.conv2_w, m.fc1_w, m.fc2_w):
        assert float(w.abs().max()) <= 0.2 + 1e-6
        assert 0.05 < float(w.std()) < 0.15
    # deterministic given seed
    m2 = LeNet5()
    torch.testing.assert_close(m.fc1_w, m2.fc1_w)


def test_mlp_forward_backward():
    torch.manual_seed(2)
    m = MLP()
    x = torch.rand(16, 28, 28, 1) - 0.5
    labels = torch.randint(0, 10, (16,))
    logits = m(x)
    assert logits.shape == (16, 10)
    loss, acc = m.loss_and_accuracy(logits, labels)
    loss.backward()
    assert m.fc1_w.grad is not None and torch.isfinite(loss)
    assert 0.0 <= float(acc) <= 1.0


def test_build_model_names():
    assert isinstance(build_model("lenet"), LeNet5)
    assert isinstance(build_model("mlp"), MLP)


def test_predictions_softmax():
    m = MLP()
    x = torch.rand(4, 28, 28, 1) - 0.5
    p = m.predictions(m(x, train=False))
    assert p.shape == (4, 10)
    torch.testing.assert_close(p.sum(1), torch.ones(4), rtol=1e-5, atol=1e-5)
