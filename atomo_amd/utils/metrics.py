"""Top-k accuracy (the helper the reference duplicates in four files —
nn_ops.py:86-99, distributed_worker.py:42-55, sync_replicas_master_nn.py:41-54,
distributed_evaluator.py:25-37 — defined once here)."""

import torch


@torch.no_grad()
def accuracy(output: torch.Tensor, target: torch.Tensor, topk=(1,)):
    maxk = max(topk)
    _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    batch = target.size(0)
    for k in topk:
        correct_k = correct[:k].reshape(-1).float().sum(0)
        res.append(float(correct_k.mul_(100.0 / batch)))
    return res
