"""accuracy() helper (the reference duplicates it in four files; defined
once here — nn_ops.py:86-99 semantics)."""

import torch

from atomo_amd.utils import accuracy


def test_topk_accuracy_exact_values():
    # logits ranking: sample 0 -> [2,1,0], 1 -> [0,2,1], 2 -> [1,0,2]
    out = torch.tensor([
        [0.1, 0.5, 0.9],
        [0.9, 0.1, 0.5],
        [0.5, 0.9, 0.1],
    ])
    target = torch.tensor([2, 2, 2])
    p1, p2, p3 = accuracy(out, target, topk=(1, 2, 3))
    assert abs(p1 - 100.0 / 3) < 1e-4   # only sample 0 top-1 correct
    assert abs(p2 - 200.0 / 3) < 1e-4   # samples 0 and 1 within top-2
    assert abs(p3 - 100.0) < 1e-4


def test_topk_accuracy_perfect_and_zero():
    out = torch.eye(4) * 5
    target = torch.arange(4)
    (p1,) = accuracy(out, target, topk=(1,))
    assert p1 == 100.0
    (z1,) = accuracy(out, (target + 1) % 4, topk=(1,))
    assert z1 == 0.0
