"""Multi-process PS loop over gloo (world_size=2, CPU) — validates the
distributed path (broadcast / gather / reduce) the GPU runs will use over
RCCL.  Spawned subprocesses rendezvous on 127.0.0.1."""

import json
import multiprocessing as mp
import os
import pickle

import pytest
import torch


def _port(base):
    """pid-offset port: avoids TIME_WAIT collisions across test runs."""
    return base + (os.getpid() % 400)


def _run_rank(rank, world, port, code, dedicated, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.set_num_threads(2)  # 2 procs share the box; avoid OpenMP thrash
    torch.manual_seed(100 + rank)
    from atomo_amd.codings import make_codec
    from atomo_amd.data import make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(backend="gloo", device=torch.device("cpu"))
    codec = make_codec(code, rank=3, quantization_level=4, bucket_size=256)
    trainer = PSTrainer(
        model_name="LeNet",
        codec=codec,
        comm=comm,
        lr=0.05,
        momentum=0.9,
        num_classes=10,
        in_channels=1,
        seed=7,
        dedicated_ps=dedicated,
        device=torch.device("cpu"),
    )
    train, _ = make_loaders("mnist", 16, 16, torch.device("cpu"), seed=50 + rank)
    losses = []
    for i, (x, y) in enumerate(train):
        losses.append(trainer.train_step(x, y))
        if i >= 9:
            break
    # weights must be identical on every rank after the final broadcast-step
    trainer.comm.broadcast(trainer.flat, src=0)
    q.put((rank, losses, trainer.flat.sum().item(), trainer.flat[:8].tolist()))
    comm.barrier()
    comm.close()


def _launch(code, dedicated, port):
    port = _port(port)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    world = 2
    procs = [
        ctx.Process(target=_run_rank, args=(r, world, port, code, dedicated, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, losses, fsum, head = q.get(timeout=180)
        results[rank] = (losses, fsum, head)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return results


@pytest.mark.parametrize("code,port", [("sgd", 29611), ("svd", 29612), ("qsgd", 29613)])
def test_dist_two_ranks(code, port):
    results = _launch(code, dedicated=False, port=port)
    assert set(results) == {0, 1}
    # all ranks converge to the same weights
    assert results[0][1] == pytest.approx(results[1][1], rel=1e-5)
    import math

    assert not math.isnan(results[0][0][-1])  # colocated master computes a loss


def test_dist_dedicated_ps():
    results = _launch("svd", dedicated=True, port=29614)
    # rank 0 is decode-only: never computes a loss
    import math

    assert all(math.isnan(l) for l in results[0][0])
    assert not any(math.isnan(l) for l in results[1][0])
    assert results[0][1] == pytest.approx(results[1][1], rel=1e-5)


def _run_rank_p2p(rank, world, port, num_agg, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.set_num_threads(2)
    torch.manual_seed(100 + rank)
    from atomo_amd.codings import make_codec
    from atomo_amd.data import make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(backend="gloo", device=torch.device("cpu"))
    code = "qsgd" if num_agg == 2 else "svd"
    trainer = PSTrainer(
        model_name="LeNet",
        codec=make_codec(code, rank=3, quantization_level=4, bucket_size=256),
        comm=comm, lr=0.05, momentum=0.9, num_classes=10, in_channels=1,
        seed=7, device=torch.device("cpu"), comm_type="P2P",
        num_aggregate=num_agg,
    )
    assert trainer.p2p
    train, _ = make_loaders("mnist", 16, 16, torch.device("cpu"), seed=50 + rank)
    losses = []
    for i, (x, y) in enumerate(train):
        losses.append(trainer.train_step(x, y))
        if i >= 7:
            break
    trainer.comm.broadcast(trainer.flat, src=0)
    q.put((rank, losses, trainer.flat.sum().item(),
           trainer._last_contrib if rank == 0 else -1))
    comm.barrier()
    comm.close()


@pytest.mark.parametrize("num_agg,port", [(0, 29621), (2, 29622)])
def test_dist_p2p_mode(num_agg, port):
    """Arrival-order P2P gather: full and partial (num_aggregate)
    aggregation both train and keep ranks weight-synchronized."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _port(port)
    world = 3
    procs = [
        ctx.Process(target=_run_rank_p2p, args=(r, world, port, num_agg, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, losses, fsum, contrib = q.get(timeout=180)
        results[rank] = (losses, fsum, contrib)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    import math

    assert not any(math.isnan(l) for l in results[0][0])
    # ranks converge to the same weights
    assert results[0][1] == pytest.approx(results[1][1], rel=1e-5)
    assert results[0][1] == pytest.approx(results[2][1], rel=1e-5)
    # contribution count: all 3 (colocated) or capped at num_aggregate
    expect = 3 if num_agg == 0 else num_agg
    assert results[0][2] == expect


def _run_rank_straggler(rank, world, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.set_num_threads(2)
    import time

    from atomo_amd.codings import make_codec
    from atomo_amd.data import make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(backend="gloo", device=torch.device("cpu"))
    trainer = PSTrainer(
        model_name="LeNet",
        codec=make_codec("svd", rank=3),
        comm=comm, lr=0.05, momentum=0.9, num_classes=10, in_channels=1,
        seed=7, device=torch.device("cpu"), comm_type="P2P", num_aggregate=2,
    )
    train, _ = make_loaders("mnist", 16, 16, torch.device("cpu"), seed=50 + rank)
    it = iter(train)
    durations = []
    for i in range(8):
        x, y = next(it)
        if rank == 2 and 2 <= i <= 5:
            time.sleep(0.4)  # persistent straggler for 4 steps
        t0 = time.perf_counter()
        trainer.train_step(x, y)
        durations.append(time.perf_counter() - t0)
    stale = getattr(comm, "stale_drops", 0)
    q.put((rank, durations, stale, trainer.flat.sum().item()))
    comm.barrier()
    comm.close()


def test_dist_partial_aggregation_drops_straggler():
    """with --num-aggregate 2 and 3 ranks, a rank that
    sleeps 400 ms per step must NOT add 400 ms to the PS's step time —
    the PS returns after the 2 fastest contributions and the late packet
    is dropped as stale."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    world, port = 3, _port(29631)
    procs = [
        ctx.Process(target=_run_rank_straggler, args=(r, world, port, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, durations, stale, fsum = q.get(timeout=240)
        results[rank] = (durations, stale, fsum)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    ps_durs = results[0][0]
    # straggler steps on the PS: well under the 400 ms sleep (full-sync
    # would absorb ~400 ms each).  Bounded staleness (2) means at most the
    # FIRST straggler step can stall the pipeline; steps 3.. must be fast.
    slow_window = ps_durs[3:6]
    assert max(slow_window) < 0.3, ps_durs
    assert results[0][1] > 0  # stale packets were observed and dropped


def _run_rank_bf16(rank, world, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.set_num_threads(2)
    from atomo_amd.codings import make_codec
    from atomo_amd.data import make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(backend="gloo", device=torch.device("cpu"))
    trainer = PSTrainer(
        model_name="LeNet", codec=make_codec("svd", rank=3), comm=comm,
        lr=0.05, momentum=0.9, num_classes=10, in_channels=1, seed=7,
        device=torch.device("cpu"), wire_dtype="bf16",
    )
    train, _ = make_loaders("mnist", 16, 16, torch.device("cpu"), seed=50 + rank)
    losses = [trainer.train_step(x, y) for _, (x, y) in zip(range(10), train)]
    trainer.comm.broadcast(trainer.flat, src=0)
    q.put((rank, losses, trainer.flat.sum().item()))
    comm.barrier()
    comm.close()


def test_dist_bf16_wire():
    """bf16 wire (weight push + svd packets): trains and keeps ranks
    weight-synchronized."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_rank_bf16, args=(r, 2, _port(29641), q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, losses, fsum = q.get(timeout=180)
        results[rank] = (losses, fsum)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    import math

    assert not any(math.isnan(l) for l in results[0][0])
    assert results[0][0][-1] < results[0][0][0] * 1.5
    assert results[0][1] == pytest.approx(results[1][1], rel=1e-5)


def test_bf16_wire_rejects_qsgd():
    from atomo_amd.codings import make_codec
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(device=torch.device("cpu"))
    with pytest.raises(ValueError):
        PSTrainer(
            model_name="LeNet", codec=make_codec("qsgd"), comm=comm,
            num_classes=10, in_channels=1, device=torch.device("cpu"),
            wire_dtype="bf16",
        )


def _run_rank_persistent(rank, world, port, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.set_num_threads(2)
    import time

    from atomo_amd.codings import make_codec
    from atomo_amd.data import make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    comm = Comm(backend="gloo", device=torch.device("cpu"))
    trainer = PSTrainer(
        model_name="LeNet", codec=make_codec("svd", rank=3), comm=comm,
        lr=0.05, momentum=0.9, num_classes=10, in_channels=1, seed=7,
        device=torch.device("cpu"), comm_type="P2P", num_aggregate=2,
    )
    train, _ = make_loaders("mnist", 16, 16, torch.device("cpu"), seed=50 + rank)
    it = iter(train)
    if rank == 2:
        # persistently slow COMPUTE (the self-skip path bypasses forward,
        # so a skipping step is fast and the rank catches up)
        trainer.model.register_forward_pre_hook(
            lambda m, inp: time.sleep(0.25)
        )
    t0 = time.perf_counter()
    for i in range(14):
        x, y = next(it)
        trainer.train_step(x, y)
    total = time.perf_counter() - t0
    q.put((rank, total, trainer.skip_count, getattr(comm, "stale_drops", 0)))
    comm.barrier()
    comm.close()


def test_dist_persistent_straggler_self_skips():
    """The tag-77 kill redesigned: a PERSISTENTLY slow rank must fall to
    the self-skip path (shipping zero packets, skipping compute) so the
    PS keeps running at the fast ranks' pace instead of throttling at
    the pipeline depth."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    world, port = 3, _port(29645)
    procs = [
        ctx.Process(target=_run_rank_persistent, args=(r, world, port, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, total, skips, stale = q.get(timeout=240)
        results[rank] = (total, skips, stale)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # 14 steps with a 250 ms/step straggler: full-sync would cost >3.5 s;
    # even a depth-6 pipeline WITHOUT self-skip throttles to ~2.2 s.
    ps_total = results[0][0]
    assert ps_total < 1.5, results
    assert results[2][1] > 0  # the slow rank actually self-skipped
