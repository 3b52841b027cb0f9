#!/usr/bin/env python3
"""Dataset preparation / verification (reference: src/data/data_prepare.py +
data_prepare.sh, which pre-download MNIST/CIFAR so all ranks start warm).

This environment has no network, so this tool (a) verifies that --data-root
holds the standard on-disk files each dataset needs (atomo_amd.data.disk
formats) and (b) otherwise reports that runs will use the synthetic data
layer.  With --warm-synthetic it pre-generates the synthetic pools once so
rank start-up is uniform."""

import argparse
import json
import sys

import torch

from atomo_amd.data import dataset_spec, make_loaders
from atomo_amd.data.disk import has_disk_data


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--data-root", type=str, default=None)
    p.add_argument("--datasets", nargs="*",
                   default=["mnist", "cifar10", "cifar100", "svhn"])
    p.add_argument("--warm-synthetic", action="store_true", default=False)
    a = p.parse_args(argv)
    report = {}
    for ds in a.datasets:
        spec = dataset_spec(ds)
        on_disk = has_disk_data(ds, a.data_root)
        report[ds] = {
            "shape": list(spec["shape"]),
            "classes": spec["classes"],
            "source": "disk" if on_disk else "synthetic",
        }
        if a.warm_synthetic and not on_disk:
            make_loaders(ds, 8, 8, torch.device("cpu"))
    print(json.dumps({"log": "data_prepare", "root": a.data_root, **report}))
    return 0


if __name__ == "__main__":
    sys.exit(main())
