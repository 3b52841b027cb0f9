"""Synthetic datasets with the reference's dataset shapes.

The reference builds torchvision MNIST/CIFAR10/SVHN/CIFAR100 loaders plus a
CIFAR-resized-to-227 "ImageNet" (distributed_nn.py:93-207) and a vendored
multiprocessing DataLoader (data_loader_ops/my_data_loader.py).  This
environment has no network for dataset downloads (BASELINE.json: synthetic
data, random-init weights), so the data layer serves random tensors of the
same shapes from a pre-generated device-resident pool — batches are sliced
on-device, which is also the right MI355X design (no host->device copies in
the hot loop).  If torchvision with local data is available the real
datasets can be swapped in via ``make_loaders(..., root=...)``.

Like the reference, there is NO sharding: every worker draws i.i.d. batches
independently, so the effective global batch is P x batch_size.
"""

from __future__ import annotations

from typing import Tuple

import torch

_SPECS = {
    "mnist": {"shape": (1, 28, 28), "classes": 10},
    "cifar10": {"shape": (3, 32, 32), "classes": 10},
    "svhn": {"shape": (3, 32, 32), "classes": 10},
    "cifar100": {"shape": (3, 32, 32), "classes": 100},
    # the reference's "imagenet" is CIFAR10 resized to 227 (distributed_nn.py:175-207)
    "imagenet": {"shape": (3, 227, 227), "classes": 10},
}


def dataset_spec(name: str) -> dict:
    if name not in _SPECS:
        raise ValueError(f"unknown dataset {name!r}; expected one of {sorted(_SPECS)}")
    return dict(_SPECS[name])


class SyntheticImageData:
    """Device-resident random pool; __iter__ yields (x, y) batches."""

    def __init__(
        self,
        dataset: str,
        batch_size: int,
        device: torch.device,
        pool_batches: int = 8,
        seed: int = 0,
        batches_per_epoch: int = 64,
    ):
        spec = dataset_spec(dataset)
        self.shape: Tuple[int, ...] = spec["shape"]
        self.classes: int = spec["classes"]
        self.batch_size = batch_size
        self.device = device
        self.batches_per_epoch = batches_per_epoch
        g = torch.Generator().manual_seed(seed)
        n = pool_batches * batch_size
        self.x = torch.randn((n, *self.shape), generator=g).to(device)
        self.y = torch.randint(0, self.classes, (n,), generator=g).to(device)
        self._pool_batches = pool_batches
        self._i = 0

    def next_batch(self):
        i = self._i % self._pool_batches
        self._i += 1
        lo, hi = i * self.batch_size, (i + 1) * self.batch_size
        return self.x[lo:hi], self.y[lo:hi]

    def __iter__(self):
        for _ in range(self.batches_per_epoch):
            yield self.next_batch()

    def __len__(self):
        return self.batches_per_epoch


def make_loaders(
    dataset: str,
    batch_size: int,
    test_batch_size: int,
    device: torch.device,
    seed: int = 0,
    root: str = None,
):
    """Synthetic loaders by default; real on-disk data when ``root`` holds
    the dataset's standard files (atomo_amd.data.disk)."""
    from .disk import DiskImageData, has_disk_data

    if has_disk_data(dataset, root):
        train = DiskImageData(dataset, root, batch_size, device, train=True,
                              seed=seed)
        test = DiskImageData(dataset, root, test_batch_size, device,
                             train=False, seed=seed + 1)
        return train, test
    train = SyntheticImageData(dataset, batch_size, device, seed=seed)
    test = SyntheticImageData(dataset, test_batch_size, device, seed=seed + 1,
                              pool_batches=4, batches_per_epoch=4)
    return train, test
