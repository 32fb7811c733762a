"""MNIST data pipeline — re-expression of /root/reference/src/mnist_data.py.

Capabilities kept: idx-gz file loading (mnist_data.py:132-154), [-0.5,0.5]
normalization (:142), DataSet with epoch shuffling + next_batch (:41-130),
fake-data mode (:60-62,102-112), read_data_sets train/validation/test split
(:156-211).  Differences by design:
  - per-rank sharding actually works (the reference threads worker_id/
    n_workers through but never uses them — every worker saw the full set);
    shard=False restores reference behavior.
  - no network download (this environment has no egress): files must exist
    locally, else use fake_data/synthetic.
  - SyntheticDataSet: device-resident random MNIST-shaped batches for
    benchmarking (no H2D copies on the hot path).
"""

from __future__ import annotations

import gzip
import os

import numpy as np
import torch

IMAGE_SIZE = 28
NUM_CHANNELS = 1
NUM_LABELS = 10
PIXEL_DEPTH = 255
VALIDATION_SIZE = 5000

TRAIN_IMAGES = "train-images-idx3-ubyte.gz"
TRAIN_LABELS = "train-labels-idx1-ubyte.gz"
TEST_IMAGES = "t10k-images-idx3-ubyte.gz"
TEST_LABELS = "t10k-labels-idx1-ubyte.gz"


def extract_images(path: str, num_images: int | None = None) -> np.ndarray:
    """idx3-ubyte.gz -> [N,28,28,1] float32 normalized to [-0.5, 0.5].

    The image count comes from the idx header (the reference hardcoded
    60000/10000; reading the header makes subsets/test fixtures loadable)."""
    import struct
    with gzip.open(path) as f:
        magic, n, rows, cols = struct.unpack(">IIII", f.read(16))
        if magic != 2051:
            raise ValueError(f"{path}: bad idx3 magic {magic}")
        if num_images is not None:
            n = min(n, num_images)
        buf = f.read(rows * cols * n * NUM_CHANNELS)
        data = np.frombuffer(buf, dtype=np.uint8).astype(np.float32)
        data = (data - PIXEL_DEPTH / 2.0) / PIXEL_DEPTH
        return data.reshape(n, rows, cols, NUM_CHANNELS)


def extract_labels(path: str, num_images: int | None = None) -> np.ndarray:
    import struct
    with gzip.open(path) as f:
        magic, n = struct.unpack(">II", f.read(8))
        if magic != 2049:
            raise ValueError(f"{path}: bad idx1 magic {magic}")
        if num_images is not None:
            n = min(n, num_images)
        buf = f.read(n)
        return np.frombuffer(buf, dtype=np.uint8).astype(np.int64)


class DataSet:
    """Epoch-shuffled batch iterator (mnist_data.py:41-130 semantics)."""

    def __init__(self, images, labels, fake_data: bool = False,
                 worker_id: int = 0, n_workers: int = 1, shard: bool = True,
                 seed: int | None = None):
        self._fake = fake_data
        if fake_data:
            self._num_examples = 10000
            self._images = None
            self._labels = None
        else:
            assert images.shape[0] == labels.shape[0]
            if shard and n_workers > 1:
                images = images[worker_id::n_workers]
                labels = labels[worker_id::n_workers]
            self._images = images
            self._labels = labels
            self._num_examples = images.shape[0]
        self._epochs_completed = 0
        self._index_in_epoch = 0
        self._rng = np.random.RandomState(seed)

    @property
    def images(self):
        return self._images

    @property
    def labels(self):
        return self._labels

    @property
    def num_examples(self):
        return self._num_examples

    @property
    def epochs_completed(self):
        return self._epochs_completed

    def next_batch(self, batch_size: int):
        if self._fake:
            image = np.zeros((IMAGE_SIZE, IMAGE_SIZE, NUM_CHANNELS), np.float32)
            label = 0
            return (np.stack([image] * batch_size),
                    np.array([label] * batch_size, dtype=np.int64))
        start = self._index_in_epoch
        self._index_in_epoch += batch_size
        if self._index_in_epoch > self._num_examples:
            self._epochs_completed += 1
            perm = self._rng.permutation(self._num_examples)
            self._images = self._images[perm]
            self._labels = self._labels[perm]
            start = 0
            self._index_in_epoch = batch_size
            assert batch_size <= self._num_examples
        end = self._index_in_epoch
        return self._images[start:end], self._labels[start:end]


class SyntheticDataSet:
    """Device-resident random MNIST-shaped data for benchmarks/tests.

    A fixed pool of `pool_size` samples is generated once on `device` in
    `dtype`; next_batch slices it round-robin — zero host involvement per
    step (the reference fed numpy feed_dicts per step through gRPC,
    distributed_train.py:310; on MI355X the batch never leaves HBM).
    """

    def __init__(self, pool_size: int = 16384, device="cpu",
                 dtype: torch.dtype = torch.float32, seed: int = 1234,
                 num_examples: int = 60000):
        g = torch.Generator(device="cpu")
        g.manual_seed(seed)
        imgs = torch.rand((pool_size, IMAGE_SIZE, IMAGE_SIZE, NUM_CHANNELS),
                          generator=g) - 0.5
        labs = torch.randint(0, NUM_LABELS, (pool_size,), generator=g)
        self._images = imgs.to(device=device, dtype=dtype)
        self._labels = labs.to(device=device)
        self._pool = pool_size
        self._pos = 0
        self._num_examples = num_examples  # nominal size for LR schedule parity
        self.epochs_completed = 0

    @property
    def num_examples(self):
        return self._num_examples

    def next_batch(self, batch_size: int):
        assert batch_size <= self._pool, "batch larger than synthetic pool"
        start = self._pos
        self._pos += batch_size
        if self._pos > self._pool:
            start, self._pos = 0, batch_size
        return (self._images[start:self._pos], self._labels[start:self._pos])


class Datasets:
    def __init__(self, train, validation, test):
        self.train = train
        self.validation = validation
        self.test = test


def read_data_sets(train_dir: str, fake_data: bool = False,
                   worker_id: int = 0, n_workers: int = 1, shard: bool = True,
                   seed: int | None = None) -> Datasets:
    if fake_data:
        return Datasets(DataSet([], [], fake_data=True),
                        DataSet([], [], fake_data=True),
                        DataSet([], [], fake_data=True))
    paths = {k: os.path.join(train_dir, v) for k, v in
             dict(ti=TRAIN_IMAGES, tl=TRAIN_LABELS, ei=TEST_IMAGES,
                  el=TEST_LABELS).items()}
    missing = [p for p in paths.values() if not os.path.exists(p)]
    if missing:
        raise FileNotFoundError(
            f"MNIST files missing under {train_dir}: {missing}. This "
            "environment has no network; place the idx-gz files there or use "
            "--subset/fake data / synthetic mode.")
    train_images = extract_images(paths["ti"])
    train_labels = extract_labels(paths["tl"])
    test_images = extract_images(paths["ei"])
    test_labels = extract_labels(paths["el"])
    # reference quirk kept: validation IS the test set (mnist_data.py:200-201)
    return Datasets(
        DataSet(train_images, train_labels, worker_id=worker_id,
                n_workers=n_workers, shard=shard, seed=seed),
        DataSet(test_images, test_labels, shard=False, seed=seed),
        DataSet(test_images, test_labels, shard=False, seed=seed))


def load_mnist(train_dir: str = "data", fake_data: bool = False,
               worker_id: int = 0, n_workers: int = 1, shard: bool = True,
               seed: int | None = None) -> Datasets:
    """Reference entry point name kept (mnist_data.py:212-213)."""
    return read_data_sets(train_dir, fake_data=fake_data, worker_id=worker_id,
                          n_workers=n_workers, shard=shard, seed=seed)
