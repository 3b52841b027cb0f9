"""Fused SGD apply (kernel: ops/csrc/atomo_kernels.hip).

One sweep over the flat parameter buffer:
    g    = grad * grad_scale + weight_decay * p
    buf  = momentum * buf + (1 - dampening) * g
    step = nesterov ? g + momentum * buf : buf        (or g if momentum == 0)
    p   -= lr * step
``grad_scale`` folds the PS's divide-by-num-workers (sync_replicas_master_nn
.py:236-239) into the same kernel, so aggregation buffers are never re-read.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import ext


def fused_sgd(
    p: torch.Tensor,
    grad: torch.Tensor,
    buf: Optional[torch.Tensor],
    lr: float,
    momentum: float = 0.0,
    weight_decay: float = 0.0,
    nesterov: bool = False,
    dampening: float = 0.0,
    grad_scale: float = 1.0,
    lr_dev: Optional[torch.Tensor] = None,
) -> None:
    """``lr_dev`` (cuda fp32 scalar) overrides ``lr`` at kernel
    execution time — under a hipGraph a captured pinned-memory copy into
    it makes lr shrinkage replay-safe (no recapture)."""
    assert p.is_cuda
    ext().fused_sgd(
        p,
        grad,
        buf if buf is not None else torch.empty(0, device=p.device),
        float(lr),
        float(momentum),
        float(weight_decay),
        bool(nesterov),
        float(dampening),
        float(grad_scale),
        lr_dev if lr_dev is not None else torch.empty(0, device=p.device),
    )
