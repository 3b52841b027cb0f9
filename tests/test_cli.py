"""Entry-point integration: single_machine in-process, distributed_nn via
torchrun subprocess (gloo, 2 ranks, 127.0.0.1), tuning parser."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_single_machine_runs(tmp_path, capsys):
    sys.path.insert(0, REPO)
    import single_machine

    rc = single_machine.main(
        [
            "--network", "LeNet", "--dataset", "mnist", "--code", "qsgd",
            "--batch-size", "8", "--max-steps", "3", "--log-interval", "1",
            "--no-cuda",
        ]
    )
    assert rc == 0
    out = capsys.readouterr().out
    recs = [json.loads(l) for l in out.splitlines() if l.startswith("{")]
    assert any(r.get("log") == "train" for r in recs)
    assert any(r.get("log") == "eval" for r in recs)


def test_tiny_tuning_parser(tmp_path):
    log = tmp_path / "run.log"
    log.write_text(
        '{"log": "train", "step": 10, "loss": 1.5}\n'
        '{"log": "train", "step": 20, "loss": 0.9}\n'
        "not json\n"
    )
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tiny_tuning_parser.py"), str(log),
         "--lr", "0.1"],
        capture_output=True, text=True, check=True,
    )
    rec = json.loads(out.stdout)
    assert rec["loss"] == 0.9 and rec["step"] == 20


@pytest.mark.timeout(300)
def test_distributed_nn_torchrun_cpu():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
            "--master-port", "29733", "distributed_nn.py",
            "--network", "LeNet", "--dataset", "mnist", "--code", "svd",
            "--svd-rank", "2", "--batch-size", "8", "--max-steps", "4",
            "--log-interval", "2", "--no-cuda",
        ],
        cwd=REPO, capture_output=True, text=True, timeout=240, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    recs = [json.loads(l) for l in out.stdout.splitlines() if l.startswith("{")]
    train = [r for r in recs if r.get("log") == "train"]
    assert train, out.stdout[-2000:]
    assert train[-1]["step"] == 4


def test_resume_latest(tmp_path, capsys):
    sys.path.insert(0, REPO)
    import distributed_nn

    args = [
        "--network", "LeNet", "--dataset", "mnist", "--code", "sgd",
        "--batch-size", "8", "--max-steps", "3", "--log-interval", "1",
        "--checkpoint-freq", "3", "--train-dir", str(tmp_path), "--no-cuda",
    ]
    assert distributed_nn.main(args) == 0
    capsys.readouterr()
    # resume and run 3 more steps: checkpoint numbering continues
    assert distributed_nn.main(args + ["--resume", "--max-steps", "6"]) == 0
    out = capsys.readouterr().out
    recs = [json.loads(l) for l in out.splitlines() if l.startswith("{")]
    res = [r for r in recs if r.get("log") == "resume"]
    assert res and res[0]["step"] == 3
    assert os.path.isfile(os.path.join(str(tmp_path), "model_step_6"))
