"""BatchedSVDEncoder (the one-sync GPU encode path) vs the per-layer oracle.
Runs on CPU — the encoder is device-agnostic; GPU runs exercise it through
test_gpu_trainer."""

import torch

from atomo_amd.codings import SVDCodec
from atomo_amd.parallel.svd_encoder import BatchedSVDEncoder


def _shapes():
    return [(16, 8, 3, 3), (64, 32, 1, 1), (10, 513), (33,), (7, 128)]


def _setup(random_sample, rank=3):
    torch.manual_seed(0)
    codec = SVDCodec(rank=rank, random_sample=random_sample,
                     generator=torch.Generator().manual_seed(9))
    grads = [torch.randn(s) for s in _shapes()]
    specs = codec.build_specs([list(g.shape) for g in grads])
    enc = BatchedSVDEncoder(codec, specs, torch.device("cpu"))
    total = sum(s.wire_words for s in specs)
    return codec, grads, specs, enc, torch.zeros(total)


def test_truncate_matches_per_layer_oracle():
    codec, grads, specs, enc, wire = _setup(random_sample=False)
    used = enc.encode_all(grads, wire)
    assert used > 0
    for g, spec in zip(grads, specs):
        region = wire[spec.wire_offset : spec.wire_offset + spec.wire_words]
        out = torch.zeros(spec.numel)
        codec.decode_from(region, out, spec)
        # reference: top-r truncated SVD reconstruction
        from atomo_amd.codings.svd import grad_to_2d

        a = grad_to_2d(g)
        u, s, vh = torch.linalg.svd(a, full_matrices=False)
        r = min(codec.rank, spec.meta["r_max"])
        best = ((u[:, :r] * s[:r]) @ vh[:r]).reshape(-1)[: spec.numel]
        assert torch.allclose(out, best, atol=1e-3), (
            spec.shape,
            (out - best).abs().max(),
        )


def test_sampled_unbiased_through_batched_path():
    codec, grads, specs, enc, wire = _setup(random_sample=True)
    acc = [torch.zeros(s.numel) for s in specs]
    n = 250
    for _ in range(n):
        enc.encode_all(grads, wire)
        for j, spec in enumerate(specs):
            region = wire[spec.wire_offset : spec.wire_offset + spec.wire_words]
            codec.decode_from(region, acc[j], spec)
    for j, (g, spec) in enumerate(zip(grads, specs)):
        rel = ((acc[j] / n) - g.reshape(-1)).norm() / g.norm()
        assert rel < 0.45, (spec.shape, rel)


def test_used_words_counted():
    codec, grads, specs, enc, wire = _setup(random_sample=False, rank=2)
    used = enc.encode_all(grads, wire)
    expect = sum(
        1 + min(2, s.meta["r_max"]) * (s.meta["m"] + s.meta["n"] + 1) for s in specs
    )
    assert used == expect
