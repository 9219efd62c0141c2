"""DistributedSampler semantics (SURVEY.md §2b row 13 contract)."""
import pytest
import torch

from pytorch_distributed_train_amd.data.sampler import DistributedSampler
from torch.utils.data import TensorDataset


def _dataset(n):
    return TensorDataset(torch.arange(n))


def test_partition_covers_all_indices():
    ds = _dataset(100)
    seen = []
    for r in range(4):
        s = DistributedSampler(ds, 4, r, shuffle=False)
        seen += list(s)
    assert sorted(seen) == list(range(100))


def test_padding_to_divisible():
    ds = _dataset(10)
    samplers = [DistributedSampler(ds, 4, r, shuffle=False) for r in range(4)]
    lens = [len(list(s)) for s in samplers]
    assert lens == [3, 3, 3, 3]  # ceil(10/4)=3, total 12 with wrap pad
    all_idx = [i for s in samplers for i in s]
    assert len(all_idx) == 12
    assert set(all_idx) == set(range(10))  # pad wraps, stays in range


def test_epoch_reshuffles_and_is_deterministic():
    ds = _dataset(64)
    s = DistributedSampler(ds, 2, 0, shuffle=True, seed=7)
    s.set_epoch(0)
    e0 = list(s)
    s.set_epoch(1)
    e1 = list(s)
    s.set_epoch(0)
    e0_again = list(s)
    assert e0 == e0_again
    assert e0 != e1


def test_ranks_disjoint_when_shuffled():
    ds = _dataset(64)
    s0 = DistributedSampler(ds, 2, 0, shuffle=True, seed=3)
    s1 = DistributedSampler(ds, 2, 1, shuffle=True, seed=3)
    s0.set_epoch(5)
    s1.set_epoch(5)
    a, b = set(s0), set(s1)
    assert a.isdisjoint(b)
    assert a | b == set(range(64))


def test_drop_last():
    ds = _dataset(10)
    s = DistributedSampler(ds, 4, 0, shuffle=False, drop_last=True)
    assert len(list(s)) == 2


def test_rank_out_of_range():
    with pytest.raises(ValueError):
        DistributedSampler(_dataset(8), 2, 2)


def test_prefetch_loader_matches_sync():
    from pytorch_distributed_train_amd.data import (DistributedSampler,
                                                    Loader, make_dataset)
    ds = make_dataset("synthetic-cifar10", 64, seed=3)
    s1 = DistributedSampler(ds, 1, 0, shuffle=True, seed=5)
    s2 = DistributedSampler(ds, 1, 0, shuffle=True, seed=5)
    s1.set_epoch(1)
    s2.set_epoch(1)
    sync = list(Loader(ds, s1, 8, workers=0))
    pref = list(Loader(ds, s2, 8, workers=4))
    assert len(sync) == len(pref)
    for (x1, y1), (x2, y2) in zip(sync, pref):
        assert torch.equal(x1, x2)
        assert torch.equal(y1, y2)


def test_process_worker_loader_matches_sync():
    """worker_mode=process (torch-DataLoader-style worker processes,
    2b#14 parity): identical batch sequence to the synchronous loader."""
    from pytorch_distributed_train_amd.data import (DistributedSampler,
                                                    Loader, make_dataset)
    import torch
    ds = make_dataset("synthetic-cifar10", 64, seed=3)
    s1 = DistributedSampler(ds, 1, 0, shuffle=True, seed=5)
    s2 = DistributedSampler(ds, 1, 0, shuffle=True, seed=5)
    sync = list(Loader(ds, s1, 8, workers=0))
    proc = list(Loader(ds, s2, 8, workers=3, worker_mode="process"))
    assert len(sync) == len(proc) > 0
    for (xa, ya), (xb, yb) in zip(sync, proc):
        assert torch.equal(xa, xb) and torch.equal(ya, yb)
