"""Learnable synthetic dataset (accuracy-evidence path, VERDICT r1
item 5): determinism, split disjointness, and actual learnability."""

import pytest
import torch

from bdbnn_amd.data import LearnableSyntheticDataset


def test_deterministic_and_split_disjoint():
    tr = LearnableSyntheticDataset(100, seed=3, split="train")
    va = LearnableSyntheticDataset(100, seed=3, split="val")
    x1, y1 = tr[5]
    x2, y2 = tr[5]
    assert torch.equal(x1, x2) and y1 == y2
    xv, _ = va[5]
    assert not torch.equal(x1, xv)  # different index stream
    # prototypes identical across splits (same generative process)
    assert torch.equal(tr.protos, va.protos)


def test_linearly_learnable_above_chance():
    """A few steps of logistic regression on flattened pixels must beat
    chance on the HELD-OUT split — i.e. the task carries real signal."""
    torch.manual_seed(0)
    tr = LearnableSyntheticDataset(512, split="train", noise=0.4,
                                   max_shift=2)
    va = LearnableSyntheticDataset(256, split="val", noise=0.4,
                                   max_shift=2)
    xs = torch.stack([tr[i][0] for i in range(512)]).flatten(1)
    ys = torch.tensor([tr[i][1] for i in range(512)])
    xv = torch.stack([va[i][0] for i in range(256)]).flatten(1)
    yv = torch.tensor([va[i][1] for i in range(256)])
    lin = torch.nn.Linear(xs.shape[1], 10)
    opt = torch.optim.Adam(lin.parameters(), lr=1e-2)
    for _ in range(60):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(lin(xs), ys)
        loss.backward()
        opt.step()
    acc = (lin(xv).argmax(1) == yv).float().mean().item()
    assert acc > 0.5, acc  # chance is 0.1


@pytest.mark.gpu
def test_short_training_learns_and_kurtosis_converges():
    """Short real training run on MI355X: resnet20 1W/1A on the
    learnable task must (a) beat chance on the held-out split and
    (b) drive layer kurtosis toward the 1.8 target."""
    import subprocess
    import sys
    import json
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = os.path.join(repo, "gpurun_out", "acc_test_log.jsonl")
    os.makedirs(os.path.dirname(out), exist_ok=True)
    # recipe calibrated on the CPU oracle path: identical settings reach
    # ~77% val top-1 there (binary nets need the higher lr to move in a
    # 10-epoch budget; lr 0.05 + noise 0.5 stays near chance on BOTH
    # paths, so a tight budget tests recipe, not kernels)
    r = subprocess.run(
        [sys.executable, "benchmarks/accuracy_run.py", "--epochs", "10",
         "--train-size", "8192", "--val-size", "1024", "--noise", "0.3",
         "--lr", "0.3", "--out", out],
        cwd=repo, capture_output=True, text=True, timeout=1200)
    assert r.returncode == 0, r.stdout + r.stderr
    summary = [json.loads(l)["summary"] for l in open(out)
               if "summary" in l][0]
    assert summary["final_val_acc1"] > 40.0, summary   # chance = 10
    assert summary["kurtosis_mean_final"] < \
        summary["kurtosis_mean_initial"] - 0.3, summary  # toward 1.8
