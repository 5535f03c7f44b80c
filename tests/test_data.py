"""Data pipeline tests: synthetic dataset, loader batching, CIFAR reader."""

import os
import pickle

import pytest

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


def _make_archive(src_root, out_path):
    import tarfile
    with tarfile.open(out_path, "w:gz") as tf:
        tf.add(os.path.join(src_root, "cifar-10-batches-py"),
               arcname="cifar-10-batches-py")


def test_download_cifar10_fetch_verify_extract(tmp_path):
    """download_cifar10 bootstraps a clean root from an archive URL with a
    checksum gate (reference surface: torchvision CIFAR10(download=True))."""
    from mi355x.data import _md5, download_cifar10

    srv = tmp_path / "srv"
    srv.mkdir()
    _write_fake_cifar(str(srv))
    tgz = tmp_path / "cifar-10-python.tar.gz"
    _make_archive(str(srv), str(tgz))
    url = "file://" + str(tgz)
    md5 = _md5(str(tgz))

    root = tmp_path / "data"
    download_cifar10(str(root), url=url, md5=md5)
    tr = CIFAR10(str(root), train=True, download=True)  # idempotent re-entry
    assert len(tr) == 40

    # checksum mismatch must refuse to extract
    bad_root = tmp_path / "bad"
    with pytest.raises(RuntimeError, match="checksum"):
        download_cifar10(str(bad_root), url=url, md5="0" * 32)


def test_download_noop_when_batches_present(tmp_path):
    """No network touch when the batches already exist on disk."""
    from mi355x.data import download_cifar10

    _write_fake_cifar(str(tmp_path))
    download_cifar10(str(tmp_path), url="http://unreachable.invalid/x.tgz")


def test_loader_accepts_num_workers():
    ds = SyntheticImageDataset(16, seed=0)
    dl = DataLoader(ds, batch_size=4, num_workers=2)
    assert sum(x.shape[0] for x, _ in dl) == 16
