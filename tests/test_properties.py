"""Property-based codec tests (hypothesis): arbitrary shapes and codec
parameters keep the wire layout consistent and the round trips bounded."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from atomo_amd.codings import QSGDCodec, SVDCodec
from atomo_amd.codings.svd import _fold2d_shape, grad_to_2d


@given(
    shape=st.lists(st.integers(1, 24), min_size=1, max_size=4),
)
@settings(max_examples=60, deadline=None)
def test_fold2d_preserves_elements(shape):
    m, n, padded = _fold2d_shape(shape)
    numel = 1
    for s in shape:
        numel *= s
    assert m * n == padded >= numel
    t = torch.arange(float(numel)).reshape(shape)
    t2 = grad_to_2d(t)
    assert t2.shape == (m, n)
    assert torch.equal(t2.reshape(-1)[:numel], t.reshape(-1))
    assert (t2.reshape(-1)[numel:] == 0).all()


@given(
    shape=st.lists(st.integers(1, 16), min_size=1, max_size=3),
    rank=st.integers(0, 5),
    sample=st.booleans(),
)
@settings(max_examples=40, deadline=None)
def test_svd_wire_region_bounds(shape, rank, sample):
    codec = SVDCodec(rank=rank, random_sample=sample,
                     generator=torch.Generator().manual_seed(0))
    g = torch.randn(shape)
    spec = codec.build_specs([list(g.shape)])[0]
    wire = torch.full((spec.wire_words,), 7.0)
    used = codec.encode_into(g, wire, spec)
    assert 0 < used <= spec.wire_words
    out = torch.zeros(spec.numel)
    codec.decode_from(wire, out, spec)
    assert torch.isfinite(out).all()


@given(
    n=st.integers(1, 700),
    q=st.integers(1, 8),
    bucket=st.sampled_from([32, 64, 256, 512]),
)
@settings(max_examples=40, deadline=None)
def test_qsgd_error_bound_property(n, q, bucket):
    codec = QSGDCodec(quantization_level=q, bucket_size=bucket,
                      generator=torch.Generator().manual_seed(1))
    g = torch.randn(n)
    out = codec.roundtrip(g)
    s = (1 << q) - 1
    nb = (n + bucket - 1) // bucket
    for b in range(nb):
        lo, hi = b * bucket, min((b + 1) * bucket, n)
        seg = torch.zeros(bucket)
        seg[: hi - lo] = g[lo:hi]
        step = seg.norm() / s
        assert (out[lo:hi] - g[lo:hi]).abs().max() <= step + 1e-5
