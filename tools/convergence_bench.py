#!/usr/bin/env python3
"""Codec-fidelity convergence benchmark that CAN fail.

The round-1 convergence study trained on a fixed synthetic pool, which a
model memorizes regardless of gradient codec.  Here the task is
teacher-student: a frozen random-init teacher labels FRESH random inputs
every step, so there is nothing to memorize — the student's loss on
incoming data is a generalization loss, and gradient-codec fidelity
directly moves the curve.  (The reference's published claim is
time-to-accuracy of SVD-rank-r vs QSGD vs vanilla SGD,
/root/reference/README.md:145-154; with no network access for CIFAR,
this is the equivalent falsifiable task on synthetic data.)

    python tools/convergence_bench.py --network ResNet18 --steps 400 \
        --codes sgd,svd,qsgd --out profiles/convergence_teacher_r18.json

Every codec run sees the identical teacher, data stream and student
init (seeded); only the gradient compression differs.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def make_teacher(network, num_classes, in_ch, device, seed):
    from atomo_amd.models import build_model

    torch.manual_seed(seed)
    t = build_model(network, num_classes, in_ch).to(device)
    t.eval()
    for p in t.parameters():
        p.requires_grad_(False)
    return t


class TeacherStream:
    """Fresh random inputs each step; labels = teacher argmax (sharpened
    by a temperature so classes are separable but non-trivial)."""

    def __init__(self, teacher, shape, batch, device, seed):
        self.teacher = teacher
        self.shape = shape
        self.batch = batch
        self.device = device
        self.gen = torch.Generator(device=device).manual_seed(seed)

    def next(self):
        x = torch.randn(
            (self.batch, *self.shape), generator=self.gen, device=self.device
        )
        with torch.no_grad():
            y = self.teacher(x).argmax(dim=1)
        return x, y


def run_code(code, a, device):
    from atomo_amd.codings import make_codec
    from atomo_amd.parallel import Comm, PSTrainer

    spec_shape = (3, 32, 32) if a.dataset == "cifar10" else (1, 28, 28)
    classes = 10
    teacher = make_teacher(a.teacher_network or a.network, classes,
                           spec_shape[0], device, seed=a.teacher_seed)
    stream = TeacherStream(teacher, spec_shape, a.batch_size, device,
                           seed=a.data_seed)
    comm = Comm(device=device)
    codec = make_codec(code, rank=a.svd_rank,
                       quantization_level=a.quantization_level,
                       bucket_size=a.bucket_size)
    trainer = PSTrainer(
        model_name=a.network, codec=codec, comm=comm, lr=a.lr,
        momentum=0.9, lr_shrinkage=a.lr_shrinkage, shrink_freq=50,
        num_classes=classes, in_channels=spec_shape[0], seed=a.seed,
        device=device, overlap=(device.type == "cuda" and code == "svd"),
    )
    losses, agree = [], []
    for step in range(a.steps):
        x, y = stream.next()
        loss = trainer.train_step(x, y)
        losses.append(float(loss))
        if (step + 1) % a.eval_freq == 0:
            xe, ye = stream.next()
            trainer.model.eval()
            with torch.no_grad():
                pred = trainer.model(xe).argmax(dim=1)
            trainer.model.train()
            agree.append(
                {"step": step + 1,
                 "teacher_agreement": float((pred == ye).float().mean())}
            )
    return {"code": code, "losses": losses, "agreement": agree}


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--network", default="ResNet18")
    p.add_argument("--teacher-network", default=None,
                   help="defaults to --network")
    p.add_argument("--dataset", default="cifar10")
    p.add_argument("--codes", default="sgd,svd,qsgd")
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--batch-size", type=int, default=128)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--lr-shrinkage", type=float, default=1.0)
    p.add_argument("--svd-rank", type=int, default=3)
    p.add_argument("--quantization-level", type=int, default=4)
    p.add_argument("--bucket-size", type=int, default=512)
    p.add_argument("--eval-freq", type=int, default=50)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--teacher-seed", type=int, default=7)
    p.add_argument("--data-seed", type=int, default=1234)
    p.add_argument("--cpu", action="store_true")
    p.add_argument("--out", default="profiles/convergence_teacher.json")
    a = p.parse_args(argv)
    device = torch.device(
        "cpu" if (a.cpu or not torch.cuda.is_available()) else "cuda:0"
    )
    results = []
    for code in a.codes.split(","):
        r = run_code(code.strip(), a, device)
        tail = sum(r["losses"][-20:]) / min(20, len(r["losses"]))
        head = sum(r["losses"][:20]) / min(20, len(r["losses"]))
        print(json.dumps({"code": r["code"], "loss_first20": head,
                          "loss_last20": tail,
                          "final_agreement": r["agreement"][-1]
                          if r["agreement"] else None}), flush=True)
        results.append(r)
    out = {
        "task": "teacher-student, fresh random batches each step "
                "(non-memorizable; loss is generalization loss)",
        "config": vars(a),
        "results": results,
    }
    os.makedirs(os.path.dirname(a.out) or ".", exist_ok=True)
    with open(a.out, "w") as f:
        json.dump(out, f)
    return 0


if __name__ == "__main__":
    sys.exit(main())
