import torch

from atomo_amd.codings.indicators import (
    gradient_indicators,
    l1_indicator,
    nuclear_indicator,
)


def test_nuclear_indicator_rank1_vs_full():
    torch.manual_seed(0)
    u, v = torch.randn(64, 1), torch.randn(1, 18)
    rank1 = u @ v
    full = torch.randn(64, 18)
    # rank-1 matrix: nuclear/fro = 1 -> indicator = 1/sqrt(min) scaled = low
    assert nuclear_indicator(rank1) < nuclear_indicator(full)


def test_l1_indicator_sparse_vs_dense():
    sparse = torch.zeros(32, 32)
    sparse[0, 0] = 5.0
    dense = torch.ones(32, 32)
    assert l1_indicator(sparse) < l1_indicator(dense)


def test_gradient_indicators_keys():
    g = torch.randn(8, 4, 3, 3)
    ind = gradient_indicators(g)
    assert set(ind) == {"nuclear", "l1", "top1_energy", "rank"}
    assert 0 < ind["top1_energy"] <= 1.0


def test_data_prepare_runs(capsys):
    import data_prepare

    rc = data_prepare.main(["--datasets", "mnist", "cifar10"])
    assert rc == 0
    import json

    rec = json.loads(capsys.readouterr().out)
    assert rec["mnist"]["source"] == "synthetic"
