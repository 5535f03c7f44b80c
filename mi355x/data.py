"""Data pipeline.

MI355X-native replacements for the torchvision/torch.utils.data slice the
reference scripts pull in (reference: cifar_example.py:38-52,
cifar_example_ddp.py:61-76 — torchvision is not installed in this image, so
these are mandatory, not convenience):

  CIFAR10                on-disk-compatible reader of the cifar-10-batches-py
                         python-pickle batches, normalized (x/255-0.5)/0.5 to
                         float32 CHW exactly like the reference's
                         transforms.Compose([ToTensor, Normalize(.5,.5)])
                         (cifar_example.py:38-40).
  SyntheticImageDataset  CIFAR-shaped random data for benchmarks (no network
                         for the real set; BASELINE.json says synthetic).
  DistributedSampler     element-for-element parity with
                         torch.utils.data.distributed.DistributedSampler
                         (rank shard, pad-by-duplication, set_epoch reseed —
                         cifar_example_ddp.py:70-76,92).
  DataLoader             batching + per-epoch reshuffle; with device= it
                         stages batches through pinned host memory and copies
                         H2D asynchronously on a dedicated HIP stream one
                         batch ahead of consumption (the MI355X replacement
                         for worker processes + pin_memory: CIFAR batches are
                         tiny, so prefetch depth 1 on a copy stream hides the
                         whole transfer behind compute).
"""

from __future__ import annotations

import math
import os
import pickle
from typing import Optional, Sequence

import numpy as np
import torch

__all__ = [
    "CIFAR10",
    "download_cifar10",
    "SyntheticImageDataset",
    "DistributedSampler",
    "DataLoader",
]


# ---------------------------------------------------------------------------
# Datasets
# ---------------------------------------------------------------------------

_CIFAR_DIR = "cifar-10-batches-py"
_TRAIN_BATCHES = [f"data_batch_{i}" for i in range(1, 6)]
_TEST_BATCHES = ["test_batch"]

CIFAR10_URL = "https://www.cs.toronto.edu/~kriz/cifar-10-python.tar.gz"
CIFAR10_MD5 = "c58f30108f718f92721af3b95e74349a"

CIFAR10_CLASSES = (
    "plane", "car", "bird", "cat", "deer",
    "dog", "frog", "horse", "ship", "truck",
)


def _md5(path: str) -> str:
    import hashlib
    h = hashlib.md5()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()


def _batches_present(base: str) -> bool:
    return all(os.path.exists(os.path.join(base, n))
               for n in _TRAIN_BATCHES + _TEST_BATCHES)


def download_cifar10(root: str, url: str = CIFAR10_URL,
                     md5: str = CIFAR10_MD5) -> None:
    """Fetch + verify + extract CIFAR-10 into the torchvision on-disk layout
    (``<root>/cifar-10-batches-py/``) — the bootstrap the reference gets
    from torchvision.datasets.CIFAR10(download=True)
    (/root/reference/cifar_example.py:40-44). No-op if the batches are
    already in place. Callers in distributed jobs must gate this on rank 0
    and barrier (consciously fixing reference quirk 11: all ranks racing on
    the download, /root/reference/cifar_example_ddp.py:67-69)."""
    import tarfile
    import urllib.request

    base = os.path.join(root, _CIFAR_DIR)
    if _batches_present(base):
        return
    os.makedirs(root, exist_ok=True)
    tgz = os.path.join(root, os.path.basename(url) or "cifar-10-python.tar.gz")
    if not (os.path.exists(tgz) and _md5(tgz) == md5):
        part = tgz + ".part"
        urllib.request.urlretrieve(url, part)
        os.replace(part, tgz)
    got = _md5(tgz)
    if got != md5:
        raise RuntimeError(
            f"CIFAR-10 archive checksum mismatch at {tgz}: got {got}, "
            f"expected {md5} — delete the file and retry")
    with tarfile.open(tgz, "r:gz") as tf:
        try:
            tf.extractall(root, filter="data")
        except TypeError:  # tarfile without extraction filters
            for m in tf.getmembers():
                if m.name.startswith(("/", "..")) or ".." in m.name.split("/"):
                    raise RuntimeError(f"unsafe path in archive: {m.name}")
            tf.extractall(root)
    if not _batches_present(base):
        raise RuntimeError(
            f"CIFAR-10 archive extracted but batches missing under {base}")


class CIFAR10:
    """Reader for the standard CIFAR-10 python-pickle batch files.

    Accepts the same on-disk layout torchvision downloads
    (``<root>/cifar-10-batches-py/data_batch_{1..5}``, ``test_batch``; each a
    pickle with ``b"data"`` as (N, 3072) uint8 row-major RRR...GGG...BBB and
    ``b"labels"`` a list of ints).  ``download=True`` bootstraps a clean box
    via :func:`download_cifar10` (reference surface:
    torchvision.datasets.CIFAR10(root, download=True),
    /root/reference/cifar_example.py:40-44).  Samples come back as float32
    (3, 32, 32) tensors normalized to [-1, 1] — identical math to the
    reference's ToTensor + Normalize((0.5,)*3, (0.5,)*3)
    (cifar_example.py:38-40).
    """

    def __init__(self, root: str, train: bool = True, download: bool = False):
        if download:
            download_cifar10(root)
        base = os.path.join(root, _CIFAR_DIR)
        names = _TRAIN_BATCHES if train else _TEST_BATCHES
        datas, labels = [], []
        for name in names:
            path = os.path.join(base, name)
            with open(path, "rb") as f:
                d = pickle.load(f, encoding="bytes")
            datas.append(np.asarray(d[b"data"], dtype=np.uint8))
            labels.extend(int(v) for v in d[b"labels"])
        raw = np.concatenate(datas, axis=0).reshape(-1, 3, 32, 32)
        # (x/255 - 0.5)/0.5 == x * (2/255) - 1 : one fused pass, float32 CHW
        self.data = torch.from_numpy(raw.astype(np.float32) * (2.0 / 255.0) - 1.0)
        self.targets = torch.tensor(labels, dtype=torch.int64)
        self.classes = list(CIFAR10_CLASSES)

    def __len__(self) -> int:
        return self.data.shape[0]

    def __getitem__(self, i: int):
        return self.data[i], self.targets[i]


class SyntheticImageDataset:
    """Random CIFAR-shaped data (float32 (3,32,32) in [-1,1], labels 0..9).

    Deterministic per seed so multi-process ranks materialize identical
    datasets without any exchange (BASELINE.json benchmarks run synthetic).
    """

    def __init__(self, n: int, seed: int = 0, size: int = 32,
                 channels: int = 3, num_classes: int = 10):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.rand((n, channels, size, size), generator=g) * 2 - 1
        self.targets = torch.randint(0, num_classes, (n,), generator=g,
                                     dtype=torch.int64)
        self.classes = [str(i) for i in range(num_classes)]

    def __len__(self) -> int:
        return self.data.shape[0]

    def __getitem__(self, i: int):
        return self.data[i], self.targets[i]


# ---------------------------------------------------------------------------
# DistributedSampler — torch-parity semantics
# ---------------------------------------------------------------------------

class DistributedSampler:
    """Rank-sharded epoch-seeded sampler, element-for-element equal to
    ``torch.utils.data.distributed.DistributedSampler`` (verified by
    tests/test_sampler.py): shuffled via ``torch.randperm`` seeded
    ``seed + epoch``, padded by duplication to a multiple of world size
    (reference behavior at cifar_example_ddp.py:70-76), strided subsample
    ``indices[rank::num_replicas]``.
    """

    def __init__(self, dataset, num_replicas: Optional[int] = None,
                 rank: Optional[int] = None, shuffle: bool = True,
                 seed: int = 0, drop_last: bool = False):
        if num_replicas is None:
            import torch.distributed as dist
            num_replicas = dist.get_world_size() if dist.is_initialized() else 1
        if rank is None:
            import torch.distributed as dist
            rank = dist.get_rank() if dist.is_initialized() else 0
        if not 0 <= rank < num_replicas:
            raise ValueError(f"rank {rank} out of range for world {num_replicas}")
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(self.dataset)
        if self.drop_last and n % self.num_replicas != 0:
            self.num_samples = math.ceil((n - self.num_replicas) / self.num_replicas)
        else:
            self.num_samples = math.ceil(n / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def __len__(self) -> int:
        return self.num_samples

    def __iter__(self):
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                if pad <= len(indices):
                    indices += indices[:pad]
                else:
                    indices += (indices * math.ceil(pad / len(indices)))[:pad]
        else:
            indices = indices[: self.total_size]
        assert len(indices) == self.total_size
        return iter(indices[self.rank: self.total_size: self.num_replicas])


# ---------------------------------------------------------------------------
# DataLoader — batching + async H2D staging
# ---------------------------------------------------------------------------

class DataLoader:
    """Minimal loader over map-style datasets with tensor ``.data``/
    ``.targets`` fast path.

    * ``shuffle=True`` reshuffles every epoch (each ``__iter__`` advances an
      internal epoch counter — matches the reference DataLoader's fresh
      permutation per epoch, cifar_example.py:46-47).
    * ``sampler=`` takes precedence over ``shuffle`` (DDP path,
      cifar_example_ddp.py:69-76); ``set_epoch`` on the sampler is the
      caller's job, as in the reference (:92).
    * ``device=`` enables prefetch: the next batch is gathered into a pinned
      staging buffer and copied H2D on a dedicated stream while the current
      batch is being consumed, then handed over with a stream-wait (no sync).
    """

    def __init__(self, dataset, batch_size: int = 1, shuffle: bool = False,
                 sampler: Optional[Sequence[int]] = None,
                 drop_last: bool = False, device=None, seed: int = 0,
                 num_workers: int = 0):
        # num_workers is accepted for reference-surface parity
        # (cifar_example.py:47,52 passes num_workers=2). Worker PROCESSES
        # are deliberately not spawned: our datasets are fully materialized
        # tensors (no per-item decode), so batch gathering is one indexed
        # copy and the GPU path overlaps H2D on a copy stream instead —
        # there is no decode work to parallelize.
        self.num_workers = num_workers
        self.dataset = dataset
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.sampler = sampler
        self.drop_last = drop_last
        self.device = torch.device(device) if device is not None else None
        self.seed = seed
        self._epoch = 0
        self._fast = hasattr(dataset, "data") and hasattr(dataset, "targets") \
            and isinstance(getattr(dataset, "data"), torch.Tensor)
        self._pinned = None  # (x, y) pinned staging buffers, lazily sized

    def __len__(self) -> int:
        if self.sampler is not None:
            n = len(self.sampler)
        else:
            n = len(self.dataset)
        if self.drop_last:
            return n // self.batch_size
        return math.ceil(n / self.batch_size)

    # -- index plan for this epoch ------------------------------------
    def _indices(self):
        if self.sampler is not None:
            return list(iter(self.sampler))
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self._epoch)
            return torch.randperm(n, generator=g).tolist()
        return list(range(n))

    def _gather(self, idx):
        if self._fast:
            t = torch.as_tensor(idx, dtype=torch.int64)
            return self.dataset.data[t], self.dataset.targets[t]
        xs, ys = zip(*(self.dataset[i] for i in idx))
        return torch.stack(xs), torch.as_tensor(ys, dtype=torch.int64)

    def __iter__(self):
        indices = self._indices()
        self._epoch += 1
        bs = self.batch_size
        batches = [indices[i: i + bs] for i in range(0, len(indices), bs)]
        if self.drop_last and batches and len(batches[-1]) < bs:
            batches.pop()
        if self.device is None or self.device.type != "cuda":
            for b in batches:
                yield self._gather(b)
            return
        yield from self._iter_device(batches)

    def _iter_device(self, batches):
        """Depth-1 pipelined H2D: gather batch k+1 into pinned staging and
        launch its copy on the copy stream while batch k computes.  Two
        pinned slots alternate; a slot is rewritten only after its previous
        H2D copy's event has completed (host-side sync on a copy that is
        already one full batch in the past, so it never actually blocks)."""
        stream = torch.cuda.Stream(device=self.device)
        if self._pinned is None:
            self._pinned = [None, None]
        slot_ev = [None, None]

        def stage(b, s):
            x, y = self._gather(b)
            if slot_ev[s] is not None:
                slot_ev[s].synchronize()  # pinned slot free for reuse?
            if self._pinned[s] is None or self._pinned[s][0].shape[0] < x.shape[0]:
                self._pinned[s] = (torch.empty_like(x).pin_memory(),
                                   torch.empty(y.shape, dtype=y.dtype).pin_memory())
            px = self._pinned[s][0][: x.shape[0]]
            py = self._pinned[s][1][: y.shape[0]]
            px.copy_(x)
            py.copy_(y)
            with torch.cuda.stream(stream):
                dx = px.to(self.device, non_blocking=True)
                dy = py.to(self.device, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(stream)
            slot_ev[s] = ev
            return dx, dy, ev

        def handover(p):
            dx, dy, ev = p
            cur = torch.cuda.current_stream(self.device)
            cur.wait_event(ev)
            # dx/dy were allocated on the copy stream; tell the caching
            # allocator they are consumed on the compute stream, else a
            # freed batch could be reused (and overwritten by a later H2D
            # copy) while compute still reads it.
            dx.record_stream(cur)
            dy.record_stream(cur)
            return dx, dy

        pending = None
        for k, b in enumerate(batches):
            nxt = stage(b, k & 1)
            if pending is not None:
                yield handover(pending)
            pending = nxt
        if pending is not None:
            yield handover(pending)
