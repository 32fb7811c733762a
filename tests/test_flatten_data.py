"""FlatParams aliasing + data pipeline tests."""

import numpy as np
import torch

from distributedmnist_amd.data import DataSet, SyntheticDataSet, load_mnist
from distributedmnist_amd.models import LeNet5
from distributedmnist_amd.parallel import FlatParams


def test_flatparams_aliasing_and_grad_accumulation():
    m = LeNet5()
    fp = FlatParams(m)
    assert fp.total == 1_663_370  # SURVEY.md section 2.4 param table
    # params are views of flat_master
    m.fc1_b.data.fill_(7.0)
    off = fp.offsets[fp.names.index("fc1_b")]
    assert float(fp.flat_master[off]) == 7.0
    # backward accumulates into flat_grad through the .grad views
    x = torch.rand(4, 28, 28, 1) - 0.5
    labels = torch.randint(0, 10, (4,))
    fp.zero_grad()
    logits = m(x, train=False)
    loss, _ = m.loss_and_accuracy(logits, labels)
    loss.backward()
    intact = fp.fix_grad_views()
    assert intact, "autograd replaced .grad views out-of-place"
    assert float(fp.flat_grad.abs().sum()) > 0


def test_flatparams_shadow_bf16():
    m = LeNet5(compute_dtype=torch.bfloat16)
    fp = FlatParams(m, compute_dtype=torch.bfloat16)
    assert fp.flat_shadow is not None
    assert m.shadows["fc1_w"].dtype == torch.bfloat16
    torch.testing.assert_close(m.shadows["fc1_w"],
                               m.fc1_w.detach().to(torch.bfloat16))
    m.fc1_w.data.add_(1.0)
    fp.sync_shadow()
    torch.testing.assert_close(m.shadows["fc1_w"],
                               m.fc1_w.detach().to(torch.bfloat16))


def test_dataset_shard_and_shuffle():
    imgs = np.arange(100, dtype=np.float32).reshape(100, 1, 1, 1)
    labs = np.arange(100, dtype=np.int64)
    d0 = DataSet(imgs, labs, worker_id=0, n_workers=2, seed=1)
    d1 = DataSet(imgs, labs, worker_id=1, n_workers=2, seed=1)
    assert d0.num_examples == 50 and d1.num_examples == 50
    s0 = set(d0.labels.tolist())
    s1 = set(d1.labels.tolist())
    assert s0.isdisjoint(s1) and len(s0 | s1) == 100
    # epoch shuffle: first epoch in order, second permuted deterministically
    b1, _ = d0.next_batch(50)
    b2, _ = d0.next_batch(50)
    assert d0.epochs_completed == 1
    assert sorted(b2.reshape(-1).tolist()) == sorted(b1.reshape(-1).tolist())
    # reference-parity mode: no sharding
    dfull = DataSet(imgs, labs, worker_id=1, n_workers=2, shard=False)
    assert dfull.num_examples == 100


def test_fake_data_mode():
    ds = load_mnist("nonexistent_dir", fake_data=True)
    x, y = ds.train.next_batch(8)
    assert x.shape == (8, 28, 28, 1) and y.shape == (8,)
    assert ds.train.num_examples == 10000


def test_synthetic_dataset():
    ds = SyntheticDataSet(pool_size=256, device="cpu", dtype=torch.float32, seed=3)
    x, y = ds.next_batch(64)
    assert x.shape == (64, 28, 28, 1)
    assert float(x.min()) >= -0.5 and float(x.max()) <= 0.5
    assert y.dtype == torch.int64 and int(y.max()) <= 9
    # wraps around the pool
    for _ in range(10):
        x2, _ = ds.next_batch(64)
    assert x2.shape == (64, 28, 28, 1)
