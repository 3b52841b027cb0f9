"""Gradient sparsity-indicator diagnostics (reference: codings/utils.py:3-8,
the fetch_indicator option in SVD.encode at codings/svd.py:97-101, and the
svd_encode diagnostic in nn_ops.py:66-82).

nuclear indicator: ||A||_* / ||A||_F * sqrt(min(m, n)) — how concentrated
the spectrum is (low = low-rank-friendly, the paper's motivation plot).
l1 indicator:      ||A||_1 / ||A||_F * sqrt(numel)   — entrywise analog.
"""

from __future__ import annotations

import math
from typing import Dict

import torch

from .svd import grad_to_2d


def nuclear_indicator(a2d: torch.Tensor, s: torch.Tensor = None) -> float:
    if s is None:
        s = torch.linalg.svdvals(a2d)
    fro = a2d.norm()
    if fro == 0:
        return 0.0
    return float(s.sum() / fro / math.sqrt(min(a2d.shape)))


def l1_indicator(a2d: torch.Tensor) -> float:
    fro = a2d.norm()
    if fro == 0:
        return 0.0
    return float(a2d.abs().sum() / fro / math.sqrt(a2d.numel()))


def gradient_indicators(grad: torch.Tensor) -> Dict[str, float]:
    a2d = grad_to_2d(grad.float())
    s = torch.linalg.svdvals(a2d)
    return {
        "nuclear": nuclear_indicator(a2d, s),
        "l1": l1_indicator(a2d),
        "top1_energy": float((s[0] ** 2) / (s**2).sum()) if s.numel() else 0.0,
        "rank": int(min(a2d.shape)),
    }
