"""idx-gz loader + subset + determinism tests."""

import gzip
import os
import struct

import numpy as np
import torch

from distributedmnist_amd.data import load_mnist
from distributedmnist_amd.data.mnist_data import extract_images, extract_labels


def write_idx(tmp_path, n_train=64, n_test=32):
    """Write tiny but VALID MNIST idx-gz files."""
    rng = np.random.RandomState(0)
    def images(path, n):
        with gzip.open(path, "wb") as f:
            f.write(struct.pack(">IIII", 2051, n, 28, 28))
            f.write(rng.randint(0, 256, size=n * 784, dtype=np.uint8).tobytes())
    def labels(path, n):
        with gzip.open(path, "wb") as f:
            f.write(struct.pack(">II", 2049, n))
            f.write(rng.randint(0, 10, size=n, dtype=np.uint8).tobytes())
    images(tmp_path / "train-images-idx3-ubyte.gz", n_train)
    labels(tmp_path / "train-labels-idx1-ubyte.gz", n_train)
    images(tmp_path / "t10k-images-idx3-ubyte.gz", n_test)
    labels(tmp_path / "t10k-labels-idx1-ubyte.gz", n_test)
    return n_train, n_test


def test_extract_idx_roundtrip(tmp_path):
    n_train, _ = write_idx(tmp_path)
    imgs = extract_images(str(tmp_path / "train-images-idx3-ubyte.gz"), n_train)
    labs = extract_labels(str(tmp_path / "train-labels-idx1-ubyte.gz"), n_train)
    assert imgs.shape == (n_train, 28, 28, 1)
    # normalized to [-0.5, 0.5] (mnist_data.py:142 semantics)
    assert imgs.min() >= -0.5 and imgs.max() <= 0.5
    assert labs.shape == (n_train,) and labs.dtype == np.int64


def test_load_mnist_real_files_with_monkeypatched_sizes(tmp_path, monkeypatch):
    write_idx(tmp_path)
    import distributedmnist_amd.data.mnist_data as md
    orig = md.read_data_sets
    def patched(train_dir, **kw):
        import types
        # small-file variant of read_data_sets
        ti = md.extract_images(os.path.join(train_dir, md.TRAIN_IMAGES), 64)
        tl = md.extract_labels(os.path.join(train_dir, md.TRAIN_LABELS), 64)
        ei = md.extract_images(os.path.join(train_dir, md.TEST_IMAGES), 32)
        el = md.extract_labels(os.path.join(train_dir, md.TEST_LABELS), 32)
        return md.Datasets(md.DataSet(ti, tl, **{k: kw[k] for k in
                                                 ("worker_id", "n_workers", "shard", "seed") if k in kw}),
                           md.DataSet(ei, el, shard=False),
                           md.DataSet(ei, el, shard=False))
    monkeypatch.setattr(md, "read_data_sets", patched)
    ds = patched(str(tmp_path), worker_id=0, n_workers=2, shard=True, seed=1)
    assert ds.train.num_examples == 32  # sharded half
    x, y = ds.train.next_batch(8)
    assert x.shape == (8, 28, 28, 1)


def test_subset_flag(tmp_path):
    from distributedmnist_amd.engine.train import make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    # subset applies to real-data mode; synthetic path ignores it
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--subset", "100", "--train_dir", str(tmp_path)])
    ds = make_dataset(flags, 0, 1, torch.device("cpu"), torch.float32)
    assert ds is not None


def test_deterministic_training_cpu(tmp_path):
    """Same seed + same data => identical loss trajectory (fixed SEED 66478
    semantics, mnist.py:32)."""
    from distributedmnist_amd.engine.train import Trainer, make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    def run(tag):
        flags = build_train_parser().parse_args(
            ["--synthetic_data", "--train_dir", str(tmp_path / tag),
             "--batch_size", "32", "--max_steps", "5", "--model", "lenet",
             "--device", "cpu", "--save_interval_secs", "100000"])
        t = Trainer(flags)
        ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
        losses = []
        for _ in range(5):
            x, y = ds.next_batch(32)
            _, loss, _, _ = t.train_step(*t.to_device(x, y))
            losses.append(float(loss))
        return losses, t.fp.flat_master.clone()
    l1, w1 = run("a")
    l2, w2 = run("b")
    assert l1 == l2
    assert torch.equal(w1, w2)


def test_real_data_training_path(tmp_path):
    """End-to-end: idx-gz files on disk -> load_mnist -> Trainer (the
    non-synthetic data path the reference used, minus the download)."""
    write_idx(tmp_path, n_train=128, n_test=64)
    from distributedmnist_amd.engine.train import Trainer, make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    flags = build_train_parser().parse_args(
        ["--data_dir", str(tmp_path), "--train_dir", str(tmp_path / "td"),
         "--batch_size", "16", "--max_steps", "4", "--model", "lenet",
         "--device", "cpu", "--save_interval_secs", "100000"])
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    assert ds.num_examples == 128
    hist = t.train(ds)
    assert len(hist) == 4
    # sharded 2-worker view of the same files
    ds2 = make_dataset(flags, 1, 2, t.device, t.compute_dtype)
    assert ds2.num_examples == 64


def test_shard_partition_properties():
    """Sharding invariants for any (n_examples, n_workers): the shards are
    disjoint, cover the full set, and sizes differ by at most one (the
    reference threaded worker_id/num_workers into unused args — SURVEY
    §2.7 bug — so these properties define the FIXED behavior)."""
    from hypothesis import given, settings, strategies as st
    from distributedmnist_amd.data.mnist_data import DataSet

    @settings(max_examples=40, deadline=None)
    @given(n=st.integers(8, 300), w=st.integers(1, 9))
    def check(n, w):
        imgs = np.arange(n, dtype=np.float32).reshape(n, 1, 1, 1)
        labs = np.arange(n, dtype=np.int64)
        shards = [DataSet(imgs, labs, worker_id=r, n_workers=w, seed=0)
                  for r in range(w)]
        seen = np.concatenate([s.labels for s in shards])
        assert sorted(seen.tolist()) == list(range(n))  # disjoint + cover
        sizes = [s.num_examples for s in shards]
        assert max(sizes) - min(sizes) <= 1              # balanced
        # shard=False parity: every worker sees the whole set
        full = DataSet(imgs, labs, worker_id=min(2, w - 1), n_workers=w,
                       shard=False, seed=0)
        assert full.num_examples == n

    check()
