"""End-to-end CLI runs (train.py main()) on tiny synthetic data —
the reference's command-line contract exercised for real."""

import os

import pytest
import torch

import train as train_mod
import bdbnn_amd.data.loaders as loaders_mod


@pytest.fixture()
def tiny_synthetic(monkeypatch):
    """Cap synthetic datasets so an epoch is a few batches."""
    orig = loaders_mod.SyntheticImageDataset

    class Small(orig):
        def __init__(self, length, *a, **kw):
            super().__init__(min(length, 64), *a, **kw)

    monkeypatch.setattr(loaders_mod, "SyntheticImageDataset", Small)
    yield


def _run_cli(tmp_path, extra):
    argv = ["./", "--dataset", "cifar10", "-a", "resnet20", "-b", "16",
            "--epochs", "1", "--synthetic-data", "--print-freq", "100",
            "--log_path", str(tmp_path), "--seed", "7", "-j", "0"] + extra
    return train_mod.main(argv)


def test_cli_train_one_epoch(tiny_synthetic, tmp_path):
    best = _run_cli(tmp_path, ["--w-kurtosis", "--weight-name", "all",
                               "--ede"])
    assert best is not None and 0.0 <= best <= 100.0
    # reference-format checkpoint written under the derived log dir
    found = []
    for root, _dirs, files in os.walk(str(tmp_path)):
        found += [f for f in files if f == "checkpoint.pth.tar"]
    assert found, "no checkpoint written"


def test_cli_evaluate_mode(tiny_synthetic, tmp_path):
    acc = _run_cli(tmp_path, ["-e"])
    assert 0.0 <= acc <= 100.0


def test_cli_resume_round_trip(tiny_synthetic, tmp_path):
    _run_cli(tmp_path / "a", [])
    ckpts = []
    for root, _dirs, files in os.walk(str(tmp_path / "a")):
        ckpts += [os.path.join(root, f) for f in files
                  if f == "checkpoint.pth.tar"]
    assert ckpts
    best = _run_cli(tmp_path / "b", ["--resume", ckpts[0],
                                     "--reset_resume"])
    assert best is not None


def test_dist_url_sets_rendezvous_env(tiny_synthetic, tmp_path, monkeypatch):
    monkeypatch.delenv("MASTER_ADDR", raising=False)
    monkeypatch.delenv("MASTER_PORT", raising=False)
    _run_cli(tmp_path, ["--dist-url", "tcp://127.0.0.1:23456", "-e"])
    assert os.environ["MASTER_ADDR"] == "127.0.0.1"
    assert os.environ["MASTER_PORT"] == "23456"
