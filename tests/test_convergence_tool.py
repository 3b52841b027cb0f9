"""Gaussian-mixture convergence benchmark: the task is non-memorizable
(fresh samples every step), so a decreasing loss is real learning and
codec fidelity can visibly separate curves."""

import json
import sys


def test_convergence_tool_runs_and_learns(tmp_path):
    sys.path.insert(0, "tools")
    from tools.convergence_bench import main

    out = tmp_path / "conv.json"
    rc = main([
        "--network", "LeNet", "--dataset", "mnist", "--codes", "sgd,svd",
        "--steps", "100", "--batch-size", "64", "--lr", "0.02",
        "--alpha", "0.3", "--eval-freq", "50", "--cpu", "--out", str(out),
    ])
    assert rc == 0
    data = json.loads(out.read_text())
    assert {r["code"] for r in data["results"]} == {"sgd", "svd"}
    for r in data["results"]:
        losses = r["losses"]
        head = sum(losses[:10]) / 10
        tail = sum(losses[-10:]) / 10
        # fresh data every step: a falling loss cannot be memorization
        assert tail < head, (r["code"], head, tail)
        assert r["holdout"][-1]["holdout_acc"] > 0.2
