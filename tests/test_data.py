"""Data pipeline tests: synthetic dataset, loader batching, CIFAR reader."""

import os
import pickle

import numpy as np
import torch

from mi355x.data import CIFAR10, DataLoader, SyntheticImageDataset


def test_synthetic_shapes_and_determinism():
    a = SyntheticImageDataset(100, seed=5)
    b = SyntheticImageDataset(100, seed=5)
    torch.testing.assert_close(a.data, b.data)
    x, y = a[3]
    assert x.shape == (3, 32, 32)
    assert 0 <= int(y) < 10


def test_loader_batches_cover_dataset():
    ds = SyntheticImageDataset(103)
    dl = DataLoader(ds, batch_size=10, shuffle=True)
    n = 0
    for x, y in dl:
        assert x.shape[0] == y.shape[0]
        n += x.shape[0]
    assert n == 103
    assert len(dl) == 11


def test_loader_shuffle_changes_per_epoch():
    ds = SyntheticImageDataset(64)
    dl = DataLoader(ds, batch_size=64, shuffle=True)
    (x1, _), = list(dl)
    (x2, _), = list(dl)
    assert not torch.equal(x1, x2)  # reseeded per epoch


def _write_fake_cifar(root):
    base = os.path.join(root, "cifar-10-batches-py")
    os.makedirs(base, exist_ok=True)
    rng = np.random.RandomState(0)
    for name, n in [("data_batch_1", 20), ("test_batch", 10)]:
        d = {b"data": rng.randint(0, 256, (n, 3072), dtype=np.uint8),
             b"labels": [int(v) for v in rng.randint(0, 10, n)]}
        with open(os.path.join(base, name), "wb") as f:
            pickle.dump(d, f)
    for i in range(2, 6):
        d = {b"data": rng.randint(0, 256, (5, 3072), dtype=np.uint8),
             b"labels": [0, 1, 2, 3, 4]}
        with open(os.path.join(base, f"data_batch_{i}"), "wb") as f:
            pickle.dump(d, f)


def test_cifar_reader_format(tmp_path):
    _write_fake_cifar(tmp_path)
    tr = CIFAR10(str(tmp_path), train=True)
    te = CIFAR10(str(tmp_path), train=False)
    assert len(tr) == 40 and len(te) == 10
    x, y = tr[0]
    assert x.shape == (3, 32, 32)
    # normalization: (x/255 - 0.5)/0.5 in [-1, 1]
    assert float(x.min()) >= -1.0 and float(x.max()) <= 1.0
    assert x.dtype == torch.float32
