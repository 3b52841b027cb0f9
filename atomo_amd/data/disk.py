"""Self-contained on-disk dataset readers (no torchvision in this stack).

The reference's data layer builds torchvision MNIST/CIFAR10/SVHN/CIFAR100
loaders plus a vendored SVHN class with md5 download (datasets.py:113-227,
distributed_nn.py:93-207).  This environment has no torchvision and no
network, so the standard on-disk binary formats are parsed directly:

  mnist     idx files (optionally .gz):  train-images-idx3-ubyte[.gz] ...
  cifar10   cifar-10-batches-py/ pickles (data_batch_1..5, test_batch)
  cifar100  cifar-100-python/ pickles (train, test)
  svhn      train_32x32.mat / test_32x32.mat (scipy.io.loadmat)

Normalization matches the reference transforms (distributed_nn.py:97,105,136):
mnist (0.1307, 0.3081); cifar10/svhn per-channel (0.4914,0.4822,0.4465)/
(0.2023,0.1994,0.2010); cifar100 125.3/123.0/113.9 means, 63.0/62.1/66.7 std.
"""

from __future__ import annotations

import gzip
import os
import pickle
import struct
from typing import Optional, Tuple

import numpy as np
import torch

_NORM = {
    "mnist": ((0.1307,), (0.3081,)),
    "cifar10": ((0.4914, 0.4822, 0.4465), (0.2023, 0.1994, 0.2010)),
    "svhn": ((0.4914, 0.4822, 0.4465), (0.2023, 0.1994, 0.2010)),
    "cifar100": (
        tuple(x / 255.0 for x in (125.3, 123.0, 113.9)),
        tuple(x / 255.0 for x in (63.0, 62.1, 66.7)),
    ),
}


def _open_maybe_gz(path: str):
    if os.path.exists(path):
        return open(path, "rb")
    if os.path.exists(path + ".gz"):
        return gzip.open(path + ".gz", "rb")
    raise FileNotFoundError(path)


def _read_idx(path: str) -> np.ndarray:
    with _open_maybe_gz(path) as f:
        magic = struct.unpack(">I", f.read(4))[0]
        ndim = magic & 0xFF
        dtype_code = (magic >> 8) & 0xFF
        assert dtype_code == 0x08, f"unsupported idx dtype {dtype_code:#x}"
        dims = struct.unpack(f">{ndim}I", f.read(4 * ndim))
        data = np.frombuffer(f.read(), dtype=np.uint8)
        return data.reshape(dims)


def load_mnist(root: str, train: bool) -> Tuple[np.ndarray, np.ndarray]:
    tag = "train" if train else "t10k"
    x = _read_idx(os.path.join(root, f"{tag}-images-idx3-ubyte"))
    y = _read_idx(os.path.join(root, f"{tag}-labels-idx1-ubyte"))
    return x[:, None, :, :].astype(np.float32) / 255.0, y.astype(np.int64)


def load_cifar10(root: str, train: bool) -> Tuple[np.ndarray, np.ndarray]:
    base = os.path.join(root, "cifar-10-batches-py")
    files = (
        [f"data_batch_{i}" for i in range(1, 6)] if train else ["test_batch"]
    )
    xs, ys = [], []
    for fn in files:
        with open(os.path.join(base, fn), "rb") as f:
            d = pickle.load(f, encoding="latin1")
        xs.append(np.asarray(d["data"], dtype=np.uint8))
        ys.extend(d["labels"])
    x = np.concatenate(xs).reshape(-1, 3, 32, 32).astype(np.float32) / 255.0
    return x, np.asarray(ys, dtype=np.int64)


def load_cifar100(root: str, train: bool) -> Tuple[np.ndarray, np.ndarray]:
    base = os.path.join(root, "cifar-100-python")
    with open(os.path.join(base, "train" if train else "test"), "rb") as f:
        d = pickle.load(f, encoding="latin1")
    x = np.asarray(d["data"], dtype=np.uint8).reshape(-1, 3, 32, 32)
    return x.astype(np.float32) / 255.0, np.asarray(d["fine_labels"], dtype=np.int64)


def load_svhn(root: str, train: bool) -> Tuple[np.ndarray, np.ndarray]:
    from scipy.io import loadmat

    mat = loadmat(os.path.join(root, f"{'train' if train else 'test'}_32x32.mat"))
    x = np.transpose(mat["X"], (3, 2, 0, 1)).astype(np.float32) / 255.0
    y = mat["y"].astype(np.int64).reshape(-1)
    y[y == 10] = 0  # SVHN labels digit '0' as 10 (reference datasets.py:113+)
    return x, y


_LOADERS = {
    "mnist": load_mnist,
    "cifar10": load_cifar10,
    "cifar100": load_cifar100,
    "svhn": load_svhn,
}


def has_disk_data(dataset: str, root: Optional[str]) -> bool:
    if not root or dataset not in _LOADERS:
        return False
    probe = {
        "mnist": ["train-images-idx3-ubyte", "train-images-idx3-ubyte.gz"],
        "cifar10": ["cifar-10-batches-py"],
        "cifar100": ["cifar-100-python"],
        "svhn": ["train_32x32.mat"],
    }[dataset]
    return any(os.path.exists(os.path.join(root, p)) for p in probe)


class DiskImageData:
    """Device-resident real dataset; same interface as SyntheticImageData.
    No sharding, independently shuffled per worker (reference semantics,
    SURVEY §2.8)."""

    def __init__(
        self,
        dataset: str,
        root: str,
        batch_size: int,
        device: torch.device,
        train: bool = True,
        seed: int = 0,
        batches_per_epoch: Optional[int] = None,
    ):
        x, y = _LOADERS[dataset](root, train)
        mean, std = _NORM[dataset]
        mean = np.asarray(mean, dtype=np.float32).reshape(1, -1, 1, 1)
        std = np.asarray(std, dtype=np.float32).reshape(1, -1, 1, 1)
        x = (x - mean) / std
        self.x = torch.from_numpy(np.ascontiguousarray(x)).to(device)
        self.y = torch.from_numpy(np.ascontiguousarray(y)).to(device)
        self.classes = int(self.y.max().item()) + 1
        self.shape = tuple(self.x.shape[1:])
        self.batch_size = batch_size
        self.device = device
        self._gen = torch.Generator().manual_seed(seed)
        n = self.x.shape[0]
        self.batches_per_epoch = batches_per_epoch or max(1, n // batch_size)
        self._perm = None
        self._i = 0

    def _reshuffle(self):
        self._perm = torch.randperm(self.x.shape[0], generator=self._gen).to(
            self.device
        )
        self._i = 0

    def next_batch(self):
        n = self.x.shape[0]
        if self._perm is None or (self._i + 1) * self.batch_size > n:
            self._reshuffle()
        lo = self._i * self.batch_size
        idx = self._perm[lo : lo + self.batch_size]
        self._i += 1
        return self.x[idx], self.y[idx]

    def __iter__(self):
        for _ in range(self.batches_per_epoch):
            yield self.next_batch()

    def __len__(self):
        return self.batches_per_epoch
