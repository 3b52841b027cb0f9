"""jacobi_eigh / jacobi_eigh_big kernels vs torch.linalg.eigh (fp64 oracle)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run_jacobi(sm, B=5, seed=None):
    from atomo_amd.ops import ext

    dev = torch.device("cuda:0")
    torch.manual_seed(seed if seed is not None else sm)
    mats = []
    grams = torch.zeros(B * sm * sm, device=dev)
    desc = torch.zeros(B, 8, dtype=torch.int64, device=dev)
    eval_offs = torch.zeros(B, dtype=torch.int64, device=dev)
    rows = torch.arange(B, dtype=torch.int32, device=dev)
    for b in range(B):
        a = torch.randn(sm + 7, sm, device=dev)
        g = a.t() @ a
        mats.append(g.clone())
        grams[b * sm * sm : (b + 1) * sm * sm] = g.reshape(-1)
        desc[b] = torch.tensor([0, sm + 7, sm, 1, b * sm * sm, 0, 0, 8])
        eval_offs[b] = b * sm
    evals = torch.zeros(B * sm, device=dev)
    if sm <= 128:
        vwarm = torch.zeros(1, device=dev)
        vw_offs = torch.zeros(B, dtype=torch.int64, device=dev)
        ext().jacobi_eigh(grams, evals, desc, eval_offs, rows, B,
                          64 if sm <= 64 else 128, vwarm, vw_offs, -1)
    else:
        v_offs = torch.tensor(
            [b * sm * sm for b in range(B)], dtype=torch.int64, device=dev
        )
        vbuf = torch.zeros(B * sm * sm, device=dev)
        ext().jacobi_eigh_big(grams, vbuf, evals, desc, eval_offs, rows, v_offs, B)
    torch.cuda.synchronize()
    return mats, grams, evals


@pytest.mark.parametrize("sm", [2, 10, 18, 63, 64, 128, 256, 512])
def test_jacobi_matches_lapack(sm):
    B = 5 if sm <= 64 else 2
    mats, grams, evals = _run_jacobi(sm, B=B)
    for b in range(B):
        g = mats[b].cpu().to(torch.float64)
        ref_vals, _ = torch.linalg.eigh(g)
        ref_vals = ref_vals.flip(0).clamp(min=0)
        got_vals = evals[b * sm : (b + 1) * sm].cpu().to(torch.float64)
        scale = max(1.0, float(ref_vals[0]))
        assert torch.allclose(got_vals, ref_vals, atol=2e-3 * scale, rtol=2e-3), (
            sm,
            (got_vals - ref_vals).abs().max(),
        )
        # eigenvectors: V^T G V diagonal with got_vals, V orthonormal
        V = grams[b * sm * sm : (b + 1) * sm * sm].view(sm, sm).cpu().to(torch.float64)
        d = V.t() @ g @ V
        off = d - torch.diag(torch.diagonal(d))
        assert off.abs().max() < 2e-2 * scale, (sm, off.abs().max())
        assert torch.allclose(
            torch.diagonal(d), got_vals, atol=2e-3 * scale, rtol=2e-3
        )
        eye = torch.eye(sm, dtype=torch.float64)
        assert (V.t() @ V - eye).abs().max() < 2e-3, sm


def test_jacobi_dense_matches_torch_eigh():
    """Dense small-batch Jacobi (one wave per matrix) vs torch.linalg.eigh
    on random symmetric batches, incl. rank-deficient and repeated-eval
    cases."""
    import torch
    from atomo_amd.ops import ext

    dev = torch.device("cuda:0")
    torch.manual_seed(5)
    for n, b in [(97, 32), (3, 64), (16, 24), (1, 8)]:
        a = torch.randn(n, b, b, device=dev)
        s = (a @ a.transpose(1, 2) / b).contiguous()
        s[0] *= 0.0  # degenerate: zero matrix
        ref = torch.linalg.eigh(s.cpu())
        ref_ev = ref.eigenvalues.flip(1)
        work = s.clone().contiguous()
        ev = torch.empty(n, b, device=dev)
        ext().jacobi_dense(work, ev, n, b)
        ev_h, v_h = ev.cpu(), work.cpu()
        scale = ref_ev.abs().max()
        assert (ev_h - ref_ev).abs().max() < 1e-4 * max(1.0, scale), (n, b)
        # eigenvector property: S v = lam v and V orthonormal
        res = s.cpu() @ v_h - v_h * ev_h.unsqueeze(1)
        assert res.abs().max() < 1e-3 * max(1.0, scale), (n, b)
        orth = v_h.transpose(1, 2) @ v_h - torch.eye(b)
        assert orth.abs().max() < 1e-3, (n, b)
