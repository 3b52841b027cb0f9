"""SVD-path GPU ops.

``gram_svd``: thin SVD of a tall-skinny gradient matrix (m >> n, n <= ~512)
via the Gram trick — G = A^T A (one rocBLAS GEMM over MFMA), eigensolve the
tiny n x n G on the host (microseconds; the sampler syncs to host for its
Bernoulli draws anyway), back-multiply U = A V / s (second GEMM).  This
inverts the reference's "SVD is faster on CPU" note (codings/svd.py:80):
on MI355X the O(m n^2) work runs on matrix cores and only the O(n^3)
eigensolve (n <= 64 for every conv layer) touches the host.

``decode_acc``: fused rank-k reconstruction u.diag(s).vT of MANY packets
accumulated into the PS aggregation buffer in ONE output sweep (kernel:
ops/csrc/atomo_kernels.hip) — W workers' packets cost one read-modify-write
of the output instead of W.
"""

from __future__ import annotations

from typing import List

import torch

from . import ext


def gram_svd(a2d: torch.Tensor, eig_floor: float = 1e-12):
    """Thin SVD (U, S, Vh) of a2d (m x n, fp32).  Columns beyond numerical
    rank get S=0 and arbitrary orthogonal directions — harmless for the
    importance sampler, which never picks p_i ~ s_i = 0 atoms."""
    m, n = a2d.shape
    if m < n:
        u, s, vh = gram_svd(a2d.t().contiguous(), eig_floor)
        return vh.t().contiguous(), s, u.t().contiguous()
    g = a2d.t() @ a2d  # n x n Gram, rocBLAS GEMM on device
    g_cpu = g.to("cpu", torch.float64)
    evals, evecs = torch.linalg.eigh(g_cpu)  # ascending
    evals = evals.flip(0).clamp(min=0.0)
    evecs = evecs.flip(1)
    s_cpu = evals.sqrt()
    v = evecs.to(a2d.device, torch.float32)
    s = s_cpu.to(a2d.device, torch.float32)
    inv_s = torch.where(s_cpu > eig_floor, 1.0 / s_cpu, torch.zeros_like(s_cpu)).to(
        a2d.device, torch.float32
    )
    u = (a2d @ v) * inv_s.unsqueeze(0)
    return u, s, v.t().contiguous()


def decode_acc(
    regions: torch.Tensor,
    out2d: torch.Tensor,
    m: int,
    n: int,
    r_max: int,
) -> None:
    """out2d (m x n fp32) += sum over packets of u.diag(s).vT.

    ``regions``: (W, wire_words) fp32 — W packets of identical layout
    [r_hat | uT (r_max x m) | s (r_max) | vT (r_max x n)].  Reads each
    packet's r_hat on-device (no host sync)."""
    assert regions.is_cuda and out2d.is_cuda
    ext().svd_decode_acc(regions, out2d, int(m), int(n), int(r_max))
