"""Our DistributedSampler vs torch's — element-for-element parity
(SURVEY.md §4 test pyramid item 4)."""

import pytest
import torch
from torch.utils.data.distributed import DistributedSampler as TorchSampler

from mi355x.data import DistributedSampler


class _Dset:
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n


@pytest.mark.parametrize("n,world", [(100, 4), (101, 4), (17, 3), (50000, 8)])
@pytest.mark.parametrize("shuffle", [True, False])
def test_parity_with_torch(n, world, shuffle):
    ds = _Dset(n)
    for rank in range(world):
        ours = DistributedSampler(ds, num_replicas=world, rank=rank,
                                  shuffle=shuffle, seed=7)
        ref = TorchSampler(ds, num_replicas=world, rank=rank,
                           shuffle=shuffle, seed=7)
        for epoch in (0, 1, 5):
            ours.set_epoch(epoch)
            ref.set_epoch(epoch)
            assert list(iter(ours)) == list(iter(ref)), (rank, epoch)


def test_partition_is_complete_and_padded():
    ds = _Dset(10)
    world = 4
    seen = []
    for rank in range(world):
        s = DistributedSampler(ds, num_replicas=world, rank=rank,
                               shuffle=False)
        idx = list(iter(s))
        assert len(idx) == 3  # ceil(10/4)
        seen += idx
    assert set(seen) == set(range(10))  # all samples covered
    assert len(seen) == 12  # padded by duplication


def test_drop_last():
    ds = _Dset(10)
    total = sum(len(list(iter(DistributedSampler(
        ds, num_replicas=4, rank=r, shuffle=False, drop_last=True))))
        for r in range(4))
    assert total == 8
