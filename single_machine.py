#!/usr/bin/env python3
"""Single-process local trainer (reference: src/single_machine.py +
nn_ops.py NN_Trainer) — no distribution, optional codec round trip in the
loop (the de-facto integration harness)."""

import json
import sys
import time

import torch

from atomo_amd.config import parse_args
from atomo_amd.data import make_loaders
from atomo_amd.parallel import Comm, PSTrainer


def main(argv=None):
    cfg = parse_args(argv)
    a = cfg.args
    device = cfg.device
    comm = Comm(device=device)  # world=1 -> no collectives
    codec = cfg.build_codec()
    trainer = PSTrainer(codec=codec, comm=comm, device=device, **cfg.trainer_kwargs())
    train_loader, test_loader = make_loaders(
        a.dataset, a.batch_size, a.test_batch_size, device, seed=a.seed,
        root=a.data_root
    )
    step = 0
    t0 = time.perf_counter()
    for epoch in range(a.epochs):
        for x, y in train_loader:
            loss = trainer.train_step(x, y)
            step += 1
            if step % a.log_interval == 0:
                print(
                    json.dumps(
                        {
                            "log": "train",
                            "epoch": epoch,
                            "step": step,
                            "loss": loss,
                            "iters_per_sec": step / (time.perf_counter() - t0),
                        }
                    ),
                    flush=True,
                )
            if step >= a.max_steps:
                ev = trainer.evaluate(test_loader)
                print(json.dumps({"log": "eval", "step": step, **ev}), flush=True)
                return 0
        ev = trainer.evaluate(test_loader)
        print(json.dumps({"log": "eval", "epoch": epoch, **ev}), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
