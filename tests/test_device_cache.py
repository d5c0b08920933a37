# -*- coding: utf-8 -*-
"""Device-resident data cache (datasets/device_cache.py): partition
round-trip, augmentation shape/statistics, determinism per seed."""
import torch

from fedtorch_amd.components.datasets.sources import (
    ArrayDataset, _cifar_train_transform)
from fedtorch_amd.components.datasets.partition import Partition
from fedtorch_amd.components.datasets.device_cache import (
    DeviceCachedLoader, NotCacheable)


def _mk(n=100, transform=None):
    torch.manual_seed(0)
    x = torch.randn(n, 3, 32, 32)
    y = torch.randint(0, 10, (n,))
    return ArrayDataset(x, y, transform=transform), x, y


def test_partition_roundtrip_and_batching():
    ds, x, y = _mk()
    ld = DeviceCachedLoader(Partition(ds, list(range(20, 80))), 16,
                            seed=5, device='cpu')
    assert len(ld) == 4
    bs = list(ld)
    assert bs[0][0].shape == (16, 3, 32, 32)
    assert bs[3][0].shape == (12, 3, 32, 32)
    tot = torch.cat([b[1] for b in bs])
    assert sorted(tot.tolist()) == sorted(y[20:80].tolist())


def test_no_transform_exact_and_sequential():
    ds, x, y = _mk()
    ld = DeviceCachedLoader(Partition(ds, list(range(10))), 4, seed=1,
                            shuffle=False, device='cpu')
    b0x, b0y = next(iter(ld))
    assert torch.equal(b0x, x[:4]) and torch.equal(b0y, y[:4])


def test_augmentation_batched():
    ds, x, y = _mk(transform=_cifar_train_transform)
    ld = DeviceCachedLoader(ds, 32, seed=3, device='cpu')
    xb, _ = next(iter(ld))
    assert xb.shape == (32, 3, 32, 32)
    # crops differ across the batch (not one shared offset)
    assert not torch.allclose(xb.std(dim=0).mean(), torch.tensor(0.0))


def test_seed_determinism():
    ds, _, _ = _mk(transform=_cifar_train_transform)
    a = [b[0].sum().item() for b in
         DeviceCachedLoader(ds, 16, seed=7, device='cpu')]
    b = [b[0].sum().item() for b in
         DeviceCachedLoader(ds, 16, seed=7, device='cpu')]
    c = [b[0].sum().item() for b in
         DeviceCachedLoader(ds, 16, seed=8, device='cpu')]
    assert a == b
    assert a != c


def test_not_cacheable_fallback():
    class Weird(torch.utils.data.Dataset):
        def __len__(self):
            return 4

        def __getitem__(self, i):
            return torch.zeros(3), 0
    try:
        DeviceCachedLoader(Weird(), 2, seed=0, device='cpu')
        raise AssertionError('should have raised')
    except NotCacheable:
        pass


def test_subset_of_partition_unwrap():
    """fed_personal splits produce Subset(Partition(ArrayDataset)) —
    the cache must compose the index chains."""
    ds, x, y = _mk()
    part = Partition(ds, list(range(10, 60)))          # base idx 10..59
    sub = torch.utils.data.Subset(part, list(range(5, 15)))  # -> 15..24
    ld = DeviceCachedLoader(sub, 4, seed=2, shuffle=False, device='cpu')
    b0x, b0y = next(iter(ld))
    assert torch.equal(b0x, x[15:19])
    assert torch.equal(b0y, y[15:19])
    assert sum(b.shape[0] for b, _ in ld) == 10
