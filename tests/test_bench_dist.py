"""Rehearse the driver's multi-GPU bench invocation on CPU: the round-end
scaling run launches bench.py under torch.distributed.run with one rank
per GPU — this test runs the same command shape (2 ranks, gloo) and
checks the preflight, the per-rank phase diagnostics and the single JSON
result line."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_under_torchrun_two_ranks():
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", "29651",
        "bench.py", "--cpu", "--network", "LeNet", "--dataset", "mnist",
        "--gpus", "2", "--steps", "3", "--warmup", "1", "--batch-size", "16",
    ]
    env = dict(os.environ)
    env["OMP_NUM_THREADS"] = "2"
    out = subprocess.run(
        cmd, cwd=REPO, env=env, capture_output=True, text=True, timeout=300
    )
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-2000:])
    # ONE result line on stdout, from rank 0
    results = [
        json.loads(line)
        for line in out.stdout.splitlines()
        if line.startswith("{") and '"metric"' in line
    ]
    assert len(results) == 1, out.stdout
    r = results[0]
    assert r["n_gpus"] == 2 and r["steps"] == 3
    assert r["value"] > 0 and r["ms_per_step"] > 0
    # whole-job aggregate: 2 colocated workers x bs 16 per step
    assert abs(r["config"]["global_batch"] - 32) < 1e-9
    # stderr carries the preflight verdict and a phase line per rank
    pre = [l for l in out.stderr.splitlines() if '"preflight"' in l]
    assert pre and json.loads(pre[0])["preflight"] == "ok"
    # the two ranks share the stderr pipe, so their records can land on
    # one line — count occurrences, not lines
    assert out.stderr.count('"comm_s"') == 2, out.stderr[-2000:]
