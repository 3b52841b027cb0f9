"""jacobi_eigh kernel vs torch.linalg.eigh (fp64 oracle)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("sm", [2, 10, 18, 63, 64])
def test_jacobi_matches_lapack(sm):
    from atomo_amd.ops import ext

    dev = torch.device("cuda:0")
    torch.manual_seed(sm)
    B = 5
    # build B PSD matrices of size sm, arranged like the encoder's buffers
    mats = []
    grams = torch.zeros(B * sm * sm, device=dev)
    desc = torch.zeros(B, 8, dtype=torch.int64, device=dev)
    eval_offs = torch.zeros(B, dtype=torch.int64, device=dev)
    for b in range(B):
        a = torch.randn(sm + 7, sm, device=dev)
        g = a.t() @ a
        mats.append(g.clone())
        grams[b * sm * sm : (b + 1) * sm * sm] = g.reshape(-1)
        # desc: [a_off, m, n, is_tall, gram_off, wire_off, stage_off, r_max]
        desc[b] = torch.tensor([0, sm + 7, sm, 1, b * sm * sm, 0, 0, 8])
        eval_offs[b] = b * sm
    evals = torch.zeros(B * sm, device=dev)
    ext().jacobi_eigh(grams, evals, desc, eval_offs, B)
    torch.cuda.synchronize()
    for b in range(B):
        g = mats[b].cpu().to(torch.float64)
        ref_vals, ref_vecs = torch.linalg.eigh(g)
        ref_vals = ref_vals.flip(0).clamp(min=0)
        got_vals = evals[b * sm : (b + 1) * sm].cpu().to(torch.float64)
        scale = max(1.0, float(ref_vals[0]))
        assert torch.allclose(got_vals, ref_vals, atol=1e-3 * scale, rtol=1e-3), (
            sm,
            (got_vals - ref_vals).abs().max(),
        )
        # eigenvectors: V^T G V should be diagonal with got_vals
        V = grams[b * sm * sm : (b + 1) * sm * sm].view(sm, sm).cpu().to(torch.float64)
        d = V.t() @ g @ V
        off = d - torch.diag(torch.diagonal(d))
        assert off.abs().max() < 1e-2 * scale, (sm, off.abs().max())
        assert torch.allclose(
            torch.diagonal(d), got_vals, atol=1e-3 * scale, rtol=1e-3
        )
        # orthonormality
        eye = torch.eye(sm, dtype=torch.float64)
        assert (V.t() @ V - eye).abs().max() < 1e-3
