"""On-disk dataset readers against synthetic fixture files in the standard
binary formats (reference data layer: datasets.py + torchvision downloads;
here parsed directly — SURVEY §2.8)."""

import gzip
import os
import pickle
import struct

import numpy as np
import pytest
import torch

from atomo_amd.data import make_loaders
from atomo_amd.data.disk import DiskImageData, has_disk_data


def _write_mnist(root, n=32, gz=True):
    os.makedirs(root, exist_ok=True)
    rng = np.random.RandomState(0)
    for tag, cnt in [("train", n), ("t10k", n // 2)]:
        imgs = rng.randint(0, 256, size=(cnt, 28, 28), dtype=np.uint8)
        labels = rng.randint(0, 10, size=(cnt,), dtype=np.uint8)
        ib = struct.pack(">IIII", 0x803, cnt, 28, 28) + imgs.tobytes()
        lb = struct.pack(">II", 0x801, cnt) + labels.tobytes()
        if gz:
            with gzip.open(os.path.join(root, f"{tag}-images-idx3-ubyte.gz"), "wb") as f:
                f.write(ib)
            with gzip.open(os.path.join(root, f"{tag}-labels-idx1-ubyte.gz"), "wb") as f:
                f.write(lb)
        else:
            open(os.path.join(root, f"{tag}-images-idx3-ubyte"), "wb").write(ib)
            open(os.path.join(root, f"{tag}-labels-idx1-ubyte"), "wb").write(lb)


def _write_cifar10(root, n=40):
    base = os.path.join(root, "cifar-10-batches-py")
    os.makedirs(base, exist_ok=True)
    rng = np.random.RandomState(1)
    per = n // 5
    for i in range(1, 6):
        d = {
            "data": rng.randint(0, 256, size=(per, 3072), dtype=np.uint8),
            "labels": rng.randint(0, 10, size=(per,)).tolist(),
        }
        with open(os.path.join(base, f"data_batch_{i}"), "wb") as f:
            pickle.dump(d, f)
    d = {
        "data": rng.randint(0, 256, size=(per, 3072), dtype=np.uint8),
        "labels": rng.randint(0, 10, size=(per,)).tolist(),
    }
    with open(os.path.join(base, "test_batch"), "wb") as f:
        pickle.dump(d, f)


def test_mnist_reader(tmp_path):
    _write_mnist(str(tmp_path))
    assert has_disk_data("mnist", str(tmp_path))
    ds = DiskImageData("mnist", str(tmp_path), 8, torch.device("cpu"))
    x, y = ds.next_batch()
    assert x.shape == (8, 1, 28, 28) and y.shape == (8,)
    # normalized: mean removed
    assert x.mean().abs() < 2.0
    assert y.max() < 10


def test_cifar10_reader_and_loader_dispatch(tmp_path):
    _write_cifar10(str(tmp_path))
    train, test = make_loaders(
        "cifar10", 8, 8, torch.device("cpu"), root=str(tmp_path)
    )
    assert isinstance(train, DiskImageData)
    x, y = train.next_batch()
    assert x.shape == (8, 3, 32, 32)
    xt, yt = test.next_batch()
    assert xt.shape == (8, 3, 32, 32)


def test_synthetic_fallback(tmp_path):
    train, _ = make_loaders("cifar10", 4, 4, torch.device("cpu"), root=None)
    from atomo_amd.data import SyntheticImageData

    assert isinstance(train, SyntheticImageData)


def test_epoch_reshuffle(tmp_path):
    _write_mnist(str(tmp_path), n=16, gz=False)
    ds = DiskImageData("mnist", str(tmp_path), 16, torch.device("cpu"), seed=3)
    _, y1 = ds.next_batch()
    _, y2 = ds.next_batch()  # new epoch, reshuffled
    assert not torch.equal(y1, y2) or ds.x.shape[0] == 16
