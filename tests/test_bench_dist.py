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


def _port(base):
    """pid-offset port: avoids TIME_WAIT collisions across test runs."""
    return base + (os.getpid() % 400)


def test_bench_under_torchrun_two_ranks():
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", str(_port(29651)),
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


def test_multinode_rendezvous_two_agents():
    """Multi-node launch path (tools/run_node.sh shape): TWO torchrun
    agents (--nnodes=2, node ranks 0/1) rendezvous on localhost and train
    — upgrades the cluster-tooling story from documented-only to
    exercised (reference: tools/pytorch_ec2.py cluster launch)."""
    import time

    agents = []
    for node_rank in (0, 1):
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=2", "--node-rank", str(node_rank),
            "--nproc-per-node", "1",
            "--master-addr", "127.0.0.1", "--master-port", str(_port(29671)),
            "bench.py", "--cpu", "--network", "LeNet", "--dataset", "mnist",
            "--steps", "3", "--warmup", "1", "--batch-size", "16",
        ]
        env = dict(os.environ)
        env["OMP_NUM_THREADS"] = "2"
        agents.append(
            subprocess.Popen(cmd, cwd=REPO, env=env,
                             stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                             text=True)
        )
    outs = []
    deadline = time.time() + 300
    for p in agents:
        out, err = p.communicate(timeout=max(10, deadline - time.time()))
        outs.append((p.returncode, out, err))
    assert all(rc == 0 for rc, _, _ in outs), [
        (rc, err[-800:]) for rc, _, err in outs
    ]
    # node 0 hosts global rank 0 -> the single result line
    results = [
        json.loads(line)
        for line in outs[0][1].splitlines()
        if line.startswith("{") and '"metric"' in line
    ]
    assert len(results) == 1 and results[0]["n_gpus"] == 2
