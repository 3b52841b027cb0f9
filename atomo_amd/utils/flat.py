"""Flat-parameter machinery.

The reference broadcasts every layer separately as fp64
(sync_replicas_master_nn.py:270-279) and keeps per-layer numpy aggregation
buffers.  MI355X-first design: all trainable parameters are VIEWS into one
contiguous fp32 device buffer, so the weight push is a single bucketed RCCL
broadcast and the optimizer apply is one fused kernel sweep.
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn


def flatten_params(model: nn.Module) -> Tuple[torch.Tensor, List[torch.Tensor]]:
    """Re-home every trainable parameter of ``model`` into one flat fp32
    buffer.  Returns (flat_buffer, list_of_param_views).  The module's
    parameters keep autograd identity (their .data becomes a view)."""
    params = [p for p in model.parameters() if p.requires_grad]
    total = sum(p.numel() for p in params)
    flat = torch.zeros(total, dtype=torch.float32, device=params[0].device)
    views = []
    off = 0
    for p in params:
        n = p.numel()
        flat[off : off + n].copy_(p.data.reshape(-1))
        p.data = flat[off : off + n].view_as(p.data)
        views.append(p.data)
        off += n
    return flat, params


def grads_of(params: List[torch.Tensor]) -> List[torch.Tensor]:
    """Gradients of the given parameters; zeros where .grad is None."""
    out = []
    for p in params:
        if p.grad is None:
            out.append(torch.zeros_like(p.data))
        else:
            out.append(p.grad)
    return out
