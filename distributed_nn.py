#!/usr/bin/env python3
"""Distributed PS training entry point (reference: src/distributed_nn.py).

Launch one process per GPU with torch.distributed (RCCL over xGMI):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 distributed_nn.py \
        --network ResNet18 --dataset cifar10 --code svd --svd-rank 3

Rank 0 is the parameter server (colocated with a worker unless
--dedicated-ps).  Role dispatch mirrors distributed_nn.py:243-260; the MPI
launcher is replaced by torchrun env-var rendezvous (RANK/WORLD_SIZE/...).
"""

import json
import sys
import time

import torch

from atomo_amd.config import parse_args
from atomo_amd.data import make_loaders
from atomo_amd.parallel import Comm, PSTrainer


def resolve_checkpoint(resume, train_dir):
    """'latest' -> newest model_step_<N> in train_dir; else the given path."""
    import os
    import re

    if resume != "latest":
        return resume if os.path.isfile(resume) else None
    best, best_n = None, -1
    if os.path.isdir(train_dir):
        for fn in os.listdir(train_dir):
            m = re.fullmatch(r"model_step_(\d+)", fn)
            if m and int(m.group(1)) > best_n:
                best_n = int(m.group(1))
                best = os.path.join(train_dir, fn)
    return best


def main(argv=None):
    cfg = parse_args(argv)
    a = cfg.args
    comm = Comm(device=None if cfg.device.type == "cuda" else cfg.device)
    device = comm.device if cfg.device.type == "cuda" else cfg.device
    codec = cfg.build_codec()
    trainer = PSTrainer(codec=codec, comm=comm, device=device, **cfg.trainer_kwargs())
    if a.resume:
        path = resolve_checkpoint(a.resume, a.train_dir)
        if path:
            trainer.load_checkpoint(path)
            if comm.rank == 0:
                print(json.dumps({"log": "resume", "path": path,
                                  "step": trainer.step_num}), flush=True)
        elif comm.rank == 0:
            print(json.dumps({"log": "resume", "path": None,
                              "note": "no checkpoint found, fresh start"}),
                  flush=True)

    train_loader, test_loader = make_loaders(
        a.dataset, a.batch_size, a.test_batch_size, device,
        seed=a.seed + comm.rank, root=a.data_root
    )

    step, done = trainer.step_num, False
    t0 = time.perf_counter()
    for epoch in range(a.epochs):
        if done:
            break
        for x, y in train_loader:
            loss = trainer.train_step(x, y)
            step += 1
            if step % a.log_interval == 0 and comm.rank == 0:
                dt = time.perf_counter() - t0
                rec = {
                    "log": "train",
                    "rank": comm.rank,
                    "epoch": epoch,
                    "step": step,
                    "loss": loss,
                    "lr": trainer.lr,
                    "iters_per_sec": step / dt,
                }
                rec.update(trainer.timers.summary())
                # wire-budget overflow: steps where the Bernoulli sample drew
                # more atoms than r_max and the tail was truncated (keeps the
                # estimator's bias surface visible; see codings/svd.py)
                if getattr(codec, "overflow_count", 0):
                    rec["svd_overflow_count"] = codec.overflow_count
                dev_bytes = trainer.wc.device_msg_bytes()
                if dev_bytes > 0:
                    # device-side Msg counter (async sampler); cumulative
                    rec["msg_bytes_device_total"] = dev_bytes
                print(json.dumps(rec), flush=True)
            if a.eval_freq and step % a.eval_freq == 0 and trainer.is_worker:
                ev = trainer.evaluate(test_loader)
                print(
                    json.dumps({"log": "eval", "rank": comm.rank, "step": step, **ev}),
                    flush=True,
                )
            if step >= a.max_steps:
                done = True
                break
    comm.barrier()
    comm.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
