"""Randomized Rayleigh-Ritz big-fold solver: subspace quality, warm-start
behavior, and the estimator-bias invariant.

The production path (BatchedSVDEncoder._solve_big_folds_randomized) is
GPU-only; these tests exercise the same math through the module-level
subspace_iterate / batched_orthonormalize helpers on CPU.  The key
semantic claim (see svd_encoder.py): the wire ships u_i = A v_i / s_i, so
a decoded atom is A v_i v_i^T / p_i — unbiased for A's projection onto
span(V) for ANY orthonormal V; eigenvalue error only moves sampling
probabilities.  Reference sampler semantics: codings/svd.py:49-67
(/root/reference/src/codings/svd.py:49-67).
"""

import torch
import pytest

from atomo_amd.parallel.svd_encoder import (
    batched_orthonormalize,
    subspace_iterate,
)


def _gradlike_gram(B, sm, decay, seed=0):
    g = torch.Generator().manual_seed(seed)
    u = torch.linalg.qr(torch.randn(B, sm, sm, generator=g)).Q
    lam = torch.exp(-torch.arange(sm, dtype=torch.float32) / decay)
    gm = u @ torch.diag_embed(lam.expand(B, sm)) @ u.transpose(1, 2)
    return 0.5 * (gm + gm.transpose(1, 2))


def _ritz(gm, b, iters, q=None, seed=1):
    if q is None:
        gen = torch.Generator().manual_seed(seed)
        q = torch.randn(gm.shape[0], gm.shape[1], b, generator=gen)
    t, q = subspace_iterate(gm, q, iters)
    ev, w = torch.linalg.eigh(t)
    return ev.flip(1).clamp(min=0.0), q @ w.flip(2)


@pytest.mark.parametrize("decay,rtol", [(2.0, 1e-2), (8.0, 1e-2), (40.0, 0.15)])
def test_topk_eigenpairs_match_exact(decay, rtol):
    # slow decay (tiny eigengaps) converges slower from cold start — the
    # steady state is covered by test_warm_start_compounds_across_steps
    gm = _gradlike_gram(3, 384, decay)
    ev, evecs = _ritz(gm, 24, iters=2)
    exact = torch.linalg.eigh(gm).eigenvalues.flip(1)
    k = 13  # r_max at svd-rank 3
    assert torch.allclose(ev[:, :k], exact[:, :k], rtol=rtol, atol=1e-4)
    # sampled (top-k) Ritz vectors are orthonormal
    vk = evecs[:, :, :k]
    eye = torch.eye(k)
    assert (vk.transpose(1, 2) @ vk - eye).abs().max() < 5e-3


def test_warm_start_compounds_across_steps():
    """Production steady state: gradients are step-correlated, so ONE
    power step per training step keeps sharpening the subspace — even the
    hardest (near-flat) spectrum converges within a few steps."""
    gm = _gradlike_gram(3, 384, 40.0)
    exact = torch.linalg.eigh(gm).eigenvalues.flip(1)
    ev, q = _ritz(gm, 24, iters=2)
    for _ in range(6):
        ev, q = _ritz(gm, 24, iters=1, q=q)
    rel = ((ev[:, :13] - exact[:, :13]) / exact[:, :13]).abs().max()
    assert rel < 0.01, float(rel)


def test_rank_deficient_gram():
    """Gradient rank <= batch size: junk directions must read eval ~ 0
    (never sampled) and the true spectrum must be recovered."""
    a = torch.randn(2, 512, 8, generator=torch.Generator().manual_seed(3))
    gm = a @ a.transpose(1, 2) / 512
    ev, evecs = _ritz(gm, 24, iters=2)
    exact = torch.linalg.eigh(gm).eigenvalues.flip(1)
    assert torch.allclose(ev[:, :8], exact[:, :8], rtol=1e-2, atol=1e-5)
    assert float(ev[:, 8:].max() / ev[:, 0].min()) < 1e-4


def test_warm_start_single_iteration_tracks_drift():
    """One warm iteration on a slightly-drifted Gram (step-to-step
    gradient correlation) matches exact eigenvalues ~as well as a cold
    2-iteration solve — the steady-state cost is ONE power step."""
    gm = _gradlike_gram(2, 256, 8.0)
    _, evecs = _ritz(gm, 24, iters=2)
    drift = torch.randn_like(gm) * 0.01 * gm.abs().max()
    gm2 = gm + 0.5 * (drift + drift.transpose(1, 2))
    ev_warm, _ = _ritz(gm2, 24, iters=1, q=evecs)
    exact = torch.linalg.eigh(gm2).eigenvalues.flip(1)
    assert torch.allclose(ev_warm[:, :13], exact[:, :13], rtol=1e-2, atol=1e-4)


def test_orthonormalize_is_span_preserving():
    y = torch.randn(2, 128, 16, generator=torch.Generator().manual_seed(5))
    q = batched_orthonormalize(y)
    # span check: projecting y onto q reproduces y
    assert torch.allclose(q @ (q.transpose(1, 2) @ y), y, atol=1e-3, rtol=1e-3)
    assert (q.transpose(1, 2) @ q - torch.eye(16)).abs().max() < 1e-4


def test_estimator_unbiased_for_subspace_projection():
    """End-to-end wire semantics with Ritz pairs: mean over many Bernoulli
    samples of sum (s_i/p_i)(A v_i/s_i) v_i^T must converge to A P_V, and
    the tail ||A - A P_V|| must be small on a decaying spectrum."""
    gen = torch.Generator().manual_seed(7)
    m, n, b, rank = 2048, 256, 24, 3
    # decaying-spectrum tall matrix (gradient-like)
    u0 = torch.linalg.qr(torch.randn(m, n, generator=gen)).Q
    v0 = torch.linalg.qr(torch.randn(n, n, generator=gen)).Q
    s0 = torch.exp(-torch.arange(n, dtype=torch.float32) / 6.0)
    a = u0 @ torch.diag(s0) @ v0.t()
    gm = (a.t() @ a).unsqueeze(0)
    ev, evecs = _ritz(gm, b, iters=2)
    s = ev[0].sqrt()
    v = evecs[0]  # (n, b)
    probs = (rank * s / s.sum()).clamp(max=1.0)
    p_v = v @ v.t()
    target = a @ p_v
    tail_rel = (a - target).norm() / a.norm()
    assert tail_rel < 0.05, f"tail energy too large: {tail_rel}"

    acc = torch.zeros_like(a)
    trials = 600
    for _ in range(trials):
        draws = torch.rand(b, generator=gen) < probs
        idx = draws.nonzero().flatten()
        if idx.numel() == 0:
            continue
        vi = v[:, idx]  # wire v^T rows
        # u_i = A v_i / s_i, shipped s = s_i / p_i -> atom = A v_i v_i^T / p_i
        acc += (a @ vi) / probs[idx] @ vi.t()
    est = acc / trials
    rel = (est - target).norm() / target.norm()
    assert rel < 0.08, f"estimator bias/variance too large: {rel}"
