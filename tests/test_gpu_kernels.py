"""HIP kernel numerics vs the CPU torch oracle (run with -m gpu on MI355X).

Every kernel is compared against the plain-PyTorch fp32 reference
implementation of the same op (atomo_amd.codings / optim)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_ext_loads(dev):
    from atomo_amd.ops import ext

    e = ext()
    assert hasattr(e, "qsgd_pack") and hasattr(e, "svd_decode_acc")


# ------------------------------------------------------------------ QSGD
@pytest.mark.parametrize("q,bucket,n", [(4, 512, 10000), (2, 256, 777), (8, 512, 512)])
def test_qsgd_gpu_pack_unpack_consistent(dev, q, bucket, n):
    """GPU pack -> CPU oracle unpack == GPU unpack (bit-exact on the wire)."""
    from atomo_amd.codings import QSGDCodec
    from atomo_amd.ops import qsgd_ops

    torch.manual_seed(0)
    g = torch.randn(n, device=dev)
    codec = QSGDCodec(quantization_level=q, bucket_size=bucket)
    spec = codec.build_specs([[n]])[0]
    wire = torch.zeros(spec.wire_words, dtype=torch.float32, device=dev)
    qsgd_ops.pack_into(g, wire, bucket, q)

    out_gpu = torch.zeros(n, device=dev)
    qsgd_ops.unpack_accumulate(wire, out_gpu, n, bucket, q)

    # CPU oracle decode of the SAME wire bytes
    wire_cpu = wire.cpu()
    out_cpu = torch.zeros(n)
    codec.decode_from(wire_cpu, out_cpu, spec)
    assert torch.allclose(out_gpu.cpu(), out_cpu, atol=1e-6), (
        (out_gpu.cpu() - out_cpu).abs().max()
    )


def test_qsgd_gpu_error_bound(dev):
    from atomo_amd.codings import QSGDCodec

    q, bucket, n = 4, 512, 4096
    torch.manual_seed(1)
    g = torch.randn(n, device=dev)
    codec = QSGDCodec(quantization_level=q, bucket_size=bucket)
    out = codec.roundtrip(g)
    for b in range(n // bucket):
        lo, hi = b * bucket, (b + 1) * bucket
        step = g[lo:hi].norm() / ((1 << q) - 1)
        assert (out[lo:hi] - g[lo:hi]).abs().max() <= step + 1e-5


def test_qsgd_gpu_unbiased(dev):
    from atomo_amd.codings import QSGDCodec

    torch.manual_seed(2)
    g = torch.randn(2048, device=dev)
    codec = QSGDCodec(quantization_level=2, bucket_size=256)
    acc = torch.zeros_like(g)
    n = 400
    for _ in range(n):
        acc += codec.roundtrip(g)
    rel = ((acc / n) - g).norm() / g.norm()
    assert rel < 0.12, rel


# ------------------------------------------------------------------ SVD decode
def test_svd_decode_acc_matches_torch(dev):
    from atomo_amd.ops import svd_ops

    torch.manual_seed(3)
    W, m, n, r_max = 4, 1000, 18, 8
    stride = 1 + r_max * (m + n + 1)
    regions = torch.zeros(W, stride, device=dev)
    expect = torch.zeros(m, n, device=dev)
    for w in range(W):
        r_hat = [3, 1, 8, 0][w]
        regions[w, 0] = float(r_hat)
        u = torch.randn(m, r_hat, device=dev)
        s = torch.rand(r_hat, device=dev) + 0.1
        vt = torch.randn(r_hat, n, device=dev)
        regions[w, 1 : 1 + r_hat * m] = u.t().reshape(-1)
        regions[w, 1 + r_max * m : 1 + r_max * m + r_hat] = s
        off = 1 + r_max * m + r_max
        regions[w, off : off + r_hat * n] = vt.reshape(-1)
        expect += (u * s) @ vt
    out = torch.randn(m, n, device=dev)  # nonzero start: kernel accumulates
    base = out.clone()
    svd_ops.decode_acc(regions, out, m, n, r_max)
    assert torch.allclose(out, base + expect, atol=1e-3), (
        (out - base - expect).abs().max()
    )


def test_svd_decode_acc_strided_view(dev):
    """regions as a narrow() of a wider stacked buffer (the gather layout)."""
    from atomo_amd.ops import svd_ops

    torch.manual_seed(4)
    W, m, n, r_max = 2, 64, 6, 4
    words = 1 + r_max * (m + n + 1)
    stacked = torch.randn(W, words + 100, device=dev)
    off0 = 37
    expect = torch.zeros(m, n, device=dev)
    for w in range(W):
        r_hat = 2
        stacked[w, off0] = float(r_hat)
        u = torch.randn(m, r_hat, device=dev)
        s = torch.rand(r_hat, device=dev)
        vt = torch.randn(r_hat, n, device=dev)
        stacked[w, off0 + 1 : off0 + 1 + r_hat * m] = u.t().reshape(-1)
        stacked[w, off0 + 1 + r_max * m : off0 + 1 + r_max * m + r_hat] = s
        o = off0 + 1 + r_max * m + r_max
        stacked[w, o : o + r_hat * n] = vt.reshape(-1)
        expect += (u * s) @ vt
    out = torch.zeros(m, n, device=dev)
    svd_ops.decode_acc(stacked.narrow(1, off0, words), out, m, n, r_max)
    assert torch.allclose(out, expect, atol=1e-4)


# ------------------------------------------------------------------ gram SVD
@pytest.mark.parametrize("m,n", [(131072, 18), (4096, 64), (500, 800)])
def test_gram_svd_reconstruction(dev, m, n):
    from atomo_amd.ops import svd_ops

    torch.manual_seed(5)
    a = torch.randn(m, n, device=dev)
    u, s, vh = svd_ops.gram_svd(a)
    k = min(m, n)
    assert u.shape == (m, k) and s.shape == (k,) and vh.shape == (k, n)
    # singular values match LAPACK
    s_ref = torch.linalg.svdvals(a.cpu())
    assert torch.allclose(s.cpu(), s_ref.float(), rtol=1e-3, atol=1e-2)
    # full reconstruction
    rec = (u * s) @ vh
    rel = (rec - a).norm() / a.norm()
    assert rel < 1e-3, rel
    # orthonormal factors
    eye = torch.eye(k, device=dev)
    assert (u.t() @ u - eye).abs().max() < 1e-2
    assert (vh @ vh.t() - eye).abs().max() < 1e-2


# ------------------------------------------------------------------ fused SGD
@pytest.mark.parametrize("momentum,nesterov,wd", [(0.9, False, 0.0),
                                                  (0.9, True, 1e-4),
                                                  (0.0, False, 0.01)])
def test_fused_sgd_matches_cpu(dev, momentum, nesterov, wd):
    from atomo_amd.optim import ExternalSGD

    torch.manual_seed(6)
    n = 100003
    p_cpu = torch.randn(n)
    p_gpu = p_cpu.to(dev)
    opt_cpu = ExternalSGD(p_cpu, lr=0.1, momentum=momentum, nesterov=nesterov,
                          weight_decay=wd)
    opt_gpu = ExternalSGD(p_gpu, lr=0.1, momentum=momentum, nesterov=nesterov,
                          weight_decay=wd)
    for _ in range(5):
        g = torch.randn(n)
        opt_cpu.step(g)
        opt_gpu.step(g.to(dev))
    assert torch.allclose(p_gpu.cpu(), p_cpu, atol=1e-5), (
        (p_gpu.cpu() - p_cpu).abs().max()
    )


def test_fused_sgd_grad_scale(dev):
    from atomo_amd.ops import optim_ops

    n = 4096
    p = torch.zeros(n, device=dev)
    g = torch.ones(n, device=dev) * 4.0
    buf = torch.zeros(n, device=dev)
    optim_ops.fused_sgd(p, g, buf, lr=1.0, momentum=0.0, grad_scale=0.25)
    assert torch.allclose(p, torch.full_like(p, -1.0))


@pytest.mark.parametrize("amsgrad", [False, True])
def test_fused_adam_matches_cpu(dev, amsgrad):
    from atomo_amd.optim import ExternalAdam

    torch.manual_seed(8)
    n = 50001
    p_cpu = torch.randn(n)
    p_gpu = p_cpu.to(dev)
    opt_cpu = ExternalAdam(p_cpu, lr=0.01, weight_decay=1e-4, amsgrad=amsgrad)
    opt_gpu = ExternalAdam(p_gpu, lr=0.01, weight_decay=1e-4, amsgrad=amsgrad)
    for _ in range(5):
        g = torch.randn(n)
        opt_cpu.step(g, grad_scale=0.5)
        opt_gpu.step(g.to(dev), grad_scale=0.5)
    assert torch.allclose(p_gpu.cpu(), p_cpu, atol=1e-5), (
        (p_gpu.cpu() - p_cpu).abs().max()
    )


def test_qsgd_terngrad_batched_gpu(dev):
    """Batched terngrad path on GPU: ternary output, bounded by clipped max."""
    from atomo_amd.codings import QSGDCodec
    from atomo_amd.parallel.wire import WireCodec

    torch.manual_seed(9)
    codec = QSGDCodec(quantization_level=1, bucket_size=256, scheme="terngrad")
    p = torch.nn.Parameter(torch.zeros(512, device=dev))
    flat = torch.randn(512, device=dev)
    p.grad = flat.view(512)
    wc = WireCodec(codec, [p], dev)
    wire = torch.zeros(wc.total_words, device=dev)
    wc.encode_all(wire, flat_grad=flat)
    agg = torch.zeros(512, device=dev)
    wc.decode_all(wire.view(1, -1), agg)
    for b in range(2):
        vals = agg[b * 256 : (b + 1) * 256]
        mx = vals.abs().max()
        uniq = vals.unique()
        for v in uniq:
            assert torch.isclose(v.abs(), mx, atol=1e-6) or v == 0


def test_fused_sgd_device_lr(dev):
    """lr_dev overrides the scalar lr at execution time (the hipGraph
    replay path for lr shrinkage)."""
    from atomo_amd.ops import optim_ops

    torch.manual_seed(3)
    p1 = torch.randn(1000, device=dev)
    p2 = p1.clone()
    g = torch.randn(1000, device=dev)
    b1 = torch.zeros(1000, device=dev)
    b2 = torch.zeros(1000, device=dev)
    optim_ops.fused_sgd(p1, g, b1, lr=0.25, momentum=0.9)
    lr_dev = torch.tensor([0.25], device=dev)
    optim_ops.fused_sgd(p2, g, b2, lr=999.0, momentum=0.9, lr_dev=lr_dev)
    assert torch.equal(p1, p2)  # scalar lr ignored when lr_dev given


def test_gpu_graph_lr_shrinkage_no_recapture(dev):
    """lr shrinkage under the whole-step graph: the SAME captured graph
    keeps replaying (no recapture) and the shrunk lr actually reaches the
    fused apply via device memory."""
    from atomo_amd.codings import make_codec
    from atomo_amd.data import make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(device=dev)
    trainer = PSTrainer(
        model_name="ResNet18", codec=make_codec("svd", rank=3), comm=comm,
        lr=0.04, momentum=0.9, lr_shrinkage=0.5, shrink_freq=3,
        num_classes=10, in_channels=3, seed=11, device=dev, use_graph=True,
    )
    assert trainer.graph_whole
    train, _ = make_loaders("cifar10", 64, 64, dev, seed=4)
    it = iter(train)
    for _ in range(4):
        x, y = next(it)
        trainer.train_step(x, y)
    g_obj = trainer._wgraph
    assert g_obj is not None
    for _ in range(8):
        x, y = next(it)
        trainer.train_step(x, y)
    assert trainer._wgraph is g_obj  # never recaptured
    assert abs(trainer.lr - 0.04 * 0.5 ** 4) < 1e-9  # 12 steps / freq 3
    assert torch.isfinite(trainer.flat).all()
    # the pinned scalar holds the lr the LAST replay applied (step 12 ran
    # with three shrinks in effect; the fourth happens post-step)
    assert abs(float(trainer._lr_host[0]) - 0.04 * 0.5 ** 3) < 1e-9
