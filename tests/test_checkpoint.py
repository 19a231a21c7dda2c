"""Checkpoint format: reference-compatible dict keys, module. prefix,
.pth.tar naming, best copy, resume/reset-resume round trip."""

import os

import torch

from bdbnn_amd.models import cifar10 as cm
from bdbnn_amd.engine.checkpoint import save_state, load_state
from bdbnn_amd.parallel import BucketedDataParallel


def _make(seed=0):
    torch.manual_seed(seed)
    model = BucketedDataParallel(cm.resnet20())
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    return model, opt


def test_checkpoint_format(tmp_path):
    model, opt = _make()
    save_state(model, opt, epoch=3, arch="resnet20", best_acc1=55.0,
               is_best=True, save_path=str(tmp_path))
    path = tmp_path / "checkpoint.pth.tar"
    assert path.exists()
    assert (tmp_path / "model_best.pth.tar").exists()
    ckpt = torch.load(str(path), map_location="cpu", weights_only=False)
    assert set(ckpt) == {"epoch", "arch", "state_dict", "best_acc1", "optimizer"}
    assert ckpt["epoch"] == 4  # ref stores epoch+1 (train.py:434)
    assert all(k.startswith("module.") for k in ckpt["state_dict"])


def test_resume_round_trip(tmp_path):
    model, opt = _make(0)
    save_state(model, opt, 5, "resnet20", 42.0, False, str(tmp_path))
    model2, opt2 = _make(1)
    start, best = load_state(str(tmp_path / "checkpoint.pth.tar"),
                             model2, opt2)
    assert start == 6 and best == 42.0
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)


def test_reset_resume_loads_weights_only(tmp_path):
    model, opt = _make(0)
    save_state(model, opt, 7, "resnet20", 42.0, False, str(tmp_path))
    model2, opt2 = _make(1)
    start, best = load_state(str(tmp_path / "checkpoint.pth.tar"),
                             model2, opt2, reset_resume=True)
    assert start == 0 and best == 0.0


def test_prefix_tolerance(tmp_path):
    model, opt = _make(0)
    save_state(model, opt, 1, "resnet20", 0.0, False, str(tmp_path))
    bare = cm.resnet20()
    load_state(str(tmp_path / "checkpoint.pth.tar"), bare)
    for (n1, p1), (n2, p2) in zip(model.module.state_dict().items(),
                                  bare.state_dict().items()):
        assert n1 == n2 and torch.equal(p1, p2)
