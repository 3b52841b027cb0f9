#!/usr/bin/env python3
"""Checkpoint-polling evaluator (reference: src/distributed_evaluator.py).

Polls --model-dir for model_step_<N> checkpoints saved by the PS
(PSTrainer.save_checkpoint) and evaluates each on a test loader; decoupled
from training through the filesystem exactly like the reference (§3.5)."""

import argparse
import json
import os
import sys
import time

import torch

from atomo_amd.data import dataset_spec, make_loaders
from atomo_amd.models import build_model
from atomo_amd.utils import accuracy


def evaluate_checkpoint(path, network, dataset, test_batch_size, device):
    spec = dataset_spec(dataset)
    model = build_model(network, spec["classes"], spec["shape"][0]).to(device)
    ckpt = torch.load(path, map_location=device, weights_only=False)
    model.load_state_dict(ckpt["model"])
    model.eval()
    _, test_loader = make_loaders(dataset, test_batch_size, test_batch_size, device)
    loss_fn = torch.nn.CrossEntropyLoss()
    tot, loss_sum, p1_sum, p5_sum = 0, 0.0, 0.0, 0.0
    with torch.no_grad():
        for x, y in test_loader:
            out = model(x)
            loss_sum += float(loss_fn(out, y)) * y.numel()
            k5 = min(5, out.shape[1])
            p1, p5 = accuracy(out, y, topk=(1, k5))
            p1_sum += p1 * y.numel()
            p5_sum += p5 * y.numel()
            tot += y.numel()
    return {
        "step": ckpt.get("step"),
        "loss": loss_sum / tot,
        "prec1": p1_sum / tot,
        "prec5": p5_sum / tot,
    }


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model-dir", type=str, default="output/models/")
    p.add_argument("--network", type=str, default="ResNet18")
    p.add_argument("--dataset", type=str, default="cifar10")
    p.add_argument("--test-batch-size", type=int, default=500)
    p.add_argument("--eval-freq", type=int, default=50,
                   help="step stride between checkpoints to look for")
    p.add_argument("--poll-interval", type=float, default=10.0)
    p.add_argument("--max-polls", type=int, default=0, help="0 = forever")
    p.add_argument("--no-cuda", action="store_true", default=False)
    a = p.parse_args(argv)
    device = torch.device(
        "cuda" if torch.cuda.is_available() and not a.no_cuda else "cpu"
    )
    next_step = a.eval_freq
    polls = 0
    while True:
        path = os.path.join(a.model_dir, f"model_step_{next_step}")
        if os.path.isfile(path):
            res = evaluate_checkpoint(
                path, a.network, a.dataset, a.test_batch_size, device
            )
            print(json.dumps({"log": "evaluator", **res}), flush=True)
            next_step += a.eval_freq
            polls = 0
        else:
            polls += 1
            if a.max_polls and polls >= a.max_polls:
                return 0
            time.sleep(a.poll_interval)


if __name__ == "__main__":
    sys.exit(main())
