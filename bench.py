#!/usr/bin/env python3
"""Flagship benchmark: ResNet-18 / CIFAR-10-shape PS training with SVD
rank-3 atomic sparsification (BASELINE.json headline config).

    python bench.py --gpus N --steps K --warmup W

For N > 1 the driver launches this via torch.distributed.run (one rank per
GPU over RCCL); rank 0 is the parameter server colocated with a worker.
Synthetic CIFAR-10-shaped data, random-init weights, fp32 compute (the
reference computes fp32; BASELINE names no lower dtype).  Reports the
whole-job aggregate images/sec plus the reference's own counters
(iters/sec, gradient MB/step/worker).
"""

import argparse
import json
import os
import time

# MIOpen kernel search (find mode 1 = Normal): ~7% faster steady-state convs
# than the default hybrid mode; the search runs during warmup steps.
os.environ.setdefault("MIOPEN_FIND_MODE", "1")

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=128)
    p.add_argument("--network", type=str, default="ResNet18")
    p.add_argument("--dataset", type=str, default="cifar10")
    p.add_argument("--code", type=str, default="svd")
    p.add_argument("--svd-rank", type=int, default=3)
    p.add_argument("--quantization-level", type=int, default=4)
    p.add_argument("--bucket-size", type=int, default=512)
    p.add_argument("--dedicated-ps", action="store_true", default=False)
    p.add_argument("--cpu", action="store_true", default=False)
    p.add_argument("--phase-log", action="store_true", default=False,
                   help="print per-phase timer breakdown to stderr")
    p.add_argument("--graph", action="store_true", default=None,
                   help="hipGraph capture (whole step at N=1 when the path "
                        "is pure device kernels; default: on for N==1)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.add_argument("--channels-last", action="store_true", default=False,
                   help="NHWC memory format for convs")
    p.add_argument("--overlap", action="store_true", default=None,
                   help="backward-hook per-layer encode on a side stream "
                        "(default: on for svd, off otherwise)")
    p.add_argument("--no-overlap", dest="overlap", action="store_false")
    p.add_argument("--amp", action="store_true", default=False,
                   help="bf16 autocast fwd/bwd (reported dtype changes)")
    p.add_argument("--wire-dtype", type=str, default="fp32",
                   choices=["fp32", "bf16"],
                   help="comm dtype for weight push + svd factor packets")
    p.add_argument("--step-times", type=str, default=None,
                   help="write per-step wall times (rank 0, ms, JSON) here; "
                        "adds a device sync per step")
    a = p.parse_args()

    from atomo_amd.codings import make_codec
    from atomo_amd.data import dataset_spec, make_loaders
    from atomo_amd.parallel import Comm, PSTrainer

    world_env = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(a.gpus, world_env)
    device = torch.device("cpu") if (a.cpu or not torch.cuda.is_available()) else None

    comm = Comm(device=device)
    device = comm.device if device is None else device
    codec = make_codec(
        a.code,
        rank=a.svd_rank,
        quantization_level=a.quantization_level,
        bucket_size=a.bucket_size,
    )
    spec = dataset_spec(a.dataset)
    trainer = PSTrainer(
        model_name=a.network,
        codec=codec,
        comm=comm,
        lr=0.01,
        momentum=0.9,
        lr_shrinkage=1.0,  # fixed lr inside the timed window
        num_classes=spec["classes"],
        in_channels=spec["shape"][0],
        dedicated_ps=a.dedicated_ps,
        seed=42,
        device=device,
        use_graph=True if a.graph is None else a.graph,
        overlap=(a.code == "svd") if a.overlap is None else a.overlap,
        defer_loss=True,
        amp=a.amp,
        wire_dtype=a.wire_dtype,
    )
    if a.channels_last:
        trainer.model.to(memory_format=torch.channels_last)
    train, _ = make_loaders(a.dataset, a.batch_size, a.batch_size, device,
                            seed=123 + comm.rank)
    if a.channels_last:
        train.x = train.x.contiguous(memory_format=torch.channels_last)

    def sync():
        comm.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    if comm.world > 1:
        # N>=2 pre-flight: verify every collective the step loop uses
        # (broadcast / gather / reduce) round-trips correctly under this
        # backend BEFORE the timed region, so an 8-GPU run fails loudly
        # in seconds rather than producing garbage
        import sys

        probe = torch.full((1024,), float(comm.rank + 1), device=device)
        bc = probe.clone()
        comm.broadcast(bc, src=0)
        ok_bc = bool((bc == 1.0).all())
        stack = (
            torch.zeros(comm.world, 1024, device=device)
            if comm.rank == 0
            else None
        )
        comm.gather(probe, stack, dst=0)
        ok_g = comm.rank != 0 or all(
            bool((stack[w] == float(w + 1)).all()) for w in range(comm.world)
        )
        red = probe.clone()
        comm.reduce_sum(red, dst=0)
        expect = sum(range(1, comm.world + 1))
        ok_r = comm.rank != 0 or bool((red == float(expect)).all())
        if not (ok_bc and ok_g and ok_r):
            print(
                json.dumps(
                    {
                        "preflight": "FAIL",
                        "rank": comm.rank,
                        "broadcast": ok_bc,
                        "gather": ok_g,
                        "reduce": ok_r,
                        "backend": comm.backend,
                    }
                ),
                file=sys.stderr,
                flush=True,
            )
            raise SystemExit(2)
        if comm.rank == 0:
            print(
                json.dumps(
                    {
                        "preflight": "ok",
                        "world": comm.world,
                        "backend": comm.backend,
                    }
                ),
                file=sys.stderr,
                flush=True,
            )

    it = iter_cycle(train)
    n_warm = a.warmup
    if device.type == "cuda" and (trainer.use_graph or trainer.graph_split):
        # graph capture happens at steps 0-1; it must land in the UNTIMED
        # region even under a tiny --warmup (reported warmup unchanged)
        n_warm = max(n_warm, 2)
    for _ in range(n_warm):
        x, y = next(it)
        trainer.train_step(x, y)

    sync()
    if comm.world > 1 and a.wire_dtype == "fp32" and not trainer.p2p:
        # weights flow from rank 0's broadcast, so after a broadcast every
        # rank must hold BITWISE-identical parameters (rank 0 runs one
        # apply ahead between steps, hence the explicit sync here); a
        # mismatch means silent comm/graph corruption — surface it before
        # the timed region
        import sys

        import torch.distributed as dist

        comm.broadcast(trainer.flat, src=0)
        sig = trainer.flat.view(torch.int32).to(torch.int64).sum()
        sigs = [torch.zeros_like(sig) for _ in range(comm.world)]
        dist.all_gather(sigs, sig)
        if comm.rank == 0:
            ok = all(bool(torch.equal(s, sigs[0])) for s in sigs)
            print(
                json.dumps(
                    {"consistency": "ok" if ok else "MISMATCH",
                     "sigs": [int(s) for s in sigs]}
                ),
                file=sys.stderr,
                flush=True,
            )
    trainer.timers.reset()
    trainer.wc.reset_device_msg_bytes()  # count timed steps only
    if a.phase_log:
        trainer.timers.sync_cuda = device.type == "cuda"
    step_ms = [] if a.step_times else None
    t0 = time.perf_counter()
    for _ in range(a.steps):
        x, y = next(it)
        if step_ms is not None:
            ts = time.perf_counter()
            trainer.train_step(x, y)
            if device.type == "cuda":
                torch.cuda.synchronize()
            step_ms.append(1e3 * (time.perf_counter() - ts))
        else:
            trainer.train_step(x, y)
    sync()
    elapsed = time.perf_counter() - t0
    if step_ms is not None and comm.rank == 0:
        qs = sorted(step_ms)
        n = len(qs)
        with open(a.step_times, "w") as f:
            json.dump(
                {
                    "steps_ms": step_ms,
                    "p5": qs[int(0.05 * n)],
                    "p50": qs[n // 2],
                    "p95": qs[min(n - 1, int(0.95 * n))],
                    "p99": qs[min(n - 1, int(0.99 * n))],
                    "max": qs[-1],
                },
                f,
            )
    if a.phase_log or comm.world > 1:
        # per-rank phase breakdown (always on for N>1: the comm phase per
        # rank is the scaling diagnostic).  Without --phase-log these are
        # host-side enqueue times for async device work.
        import sys

        rec = {"rank": comm.rank, **trainer.timers.summary()}
        sd = getattr(comm, "stale_drops", 0)
        if sd:
            rec["stale_drops"] = sd
        print(json.dumps(rec), file=sys.stderr, flush=True)

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if comm.backend == "nccl" else "cpu")
    if comm.world > 1:
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    n_workers = trainer.num_workers
    iters_per_sec = a.steps / elapsed
    images_per_sec = iters_per_sec * a.batch_size * n_workers
    msg_bytes = trainer.timers.scalars.get("msg_bytes", 0.0)
    steps_counted = max(1, trainer.timers.counts.get("msg_bytes", 1))
    # both counters cover the timed steps only (device counter reset after
    # warmup); host scalar and device counter partition the layers
    dev_bytes = trainer.wc.device_msg_bytes()
    grad_mb_per_step = (msg_bytes / steps_counted + dev_bytes / a.steps) / 1e6

    if comm.rank == 0:
        print(
            json.dumps(
                {
                    "metric": (
                        "images/sec (ResNet-18 SVD-r=3 PS data-parallel)"
                        if (a.network, a.code, a.svd_rank)
                        == ("ResNet18", "svd", 3)
                        else f"images/sec ({a.network} {a.code}"
                        + (f"-r{a.svd_rank}" if a.code == "svd" else "")
                        + " PS data-parallel)"
                    ),
                    "value": images_per_sec,
                    "unit": "images/s",
                    "n_gpus": n_gpus,
                    "steps": a.steps,
                    "warmup": a.warmup,
                    "ms_per_step": 1e3 * elapsed / a.steps,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16-autocast" if a.amp else "fp32",
                    "data": "synthetic (CIFAR-10-shaped random, random-init weights)",
                    "iters_per_sec": iters_per_sec,
                    "grad_mb_per_step_per_worker": grad_mb_per_step,
                    "config": {
                        "model": a.network,
                        "global_batch": a.batch_size * n_workers,
                        "batch_per_worker": a.batch_size,
                        "input": list(spec["shape"]),
                        "code": a.code,
                        "svd_rank": a.svd_rank,
                        "parallelism": f"ps-dp{n_gpus}"
                        + ("-dedicated" if a.dedicated_ps else "-colocated"),
                    },
                }
            ),
            flush=True,
        )
    comm.barrier()
    comm.close()


def iter_cycle(loader):
    while True:
        for batch in loader:
            yield batch


if __name__ == "__main__":
    main()
