"""One-launch batched QSGD (descriptor table) vs the CPU oracle."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _build(dev, shapes, q=4, bucket=512):
    from atomo_amd.codings import QSGDCodec
    from atomo_amd.parallel.wire import WireCodec

    codec = QSGDCodec(quantization_level=q, bucket_size=bucket)
    numels = [int(torch.Size(s).numel()) for s in shapes]
    total = sum(numels)
    flat = torch.randn(total, device=dev)
    params = []
    off = 0
    for s, n in zip(shapes, numels):
        p = torch.nn.Parameter(torch.zeros(s, device=dev))
        p.grad = flat[off : off + n].view(s)
        params.append(p)
        off += n
    wc = WireCodec(codec, params, dev)
    return codec, wc, flat


def test_batched_qsgd_roundtrip_matches_oracle():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    shapes = [(64, 16, 3, 3), (513,), (10, 512), (100,)]
    codec, wc, flat = _build(dev, shapes)
    assert wc._qsgd_tables is not None
    wire = torch.zeros(wc.total_words, device=dev)
    used = wc.encode_all(wire, flat_grad=flat)
    assert used == wc.total_words

    # GPU decode
    agg_gpu = torch.zeros_like(flat)
    wc.decode_all(wire.view(1, -1), agg_gpu)

    # CPU oracle decode of the same wire bytes
    wire_cpu = wire.cpu()
    agg_cpu = torch.zeros(flat.numel())
    for spec, p_off in zip(wc.specs, wc.param_offsets):
        region = wire_cpu[spec.wire_offset : spec.wire_offset + spec.wire_words]
        codec.decode_from(region, agg_cpu[p_off : p_off + spec.numel], spec)
    assert torch.allclose(agg_gpu.cpu(), agg_cpu, atol=1e-6), (
        (agg_gpu.cpu() - agg_cpu).abs().max()
    )
    # error bound per element: one quantization step of its bucket
    err = (agg_gpu - flat).abs().max().item()
    assert math.isfinite(err)


def test_batched_qsgd_unbiased():
    dev = torch.device("cuda:0")
    torch.manual_seed(1)
    shapes = [(32, 8, 3, 3), (1000,)]
    codec, wc, flat = _build(dev, shapes, q=2, bucket=256)
    wire = torch.zeros(wc.total_words, device=dev)
    acc = torch.zeros_like(flat)
    n = 300
    for _ in range(n):
        wc.encode_all(wire, flat_grad=flat)
        wc.decode_all(wire.view(1, -1), acc)
    rel = ((acc / n) - flat).norm() / flat.norm()
    assert rel < 0.12, rel
