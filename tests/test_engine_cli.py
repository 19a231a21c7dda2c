"""Engine + CLI integration: flags parse, EDE injection, kurtosis table,
a short CPU training run decreases loss, evaluate mode, TS path."""

import torch
import pytest

from train import build_parser, build_model, build_teacher
from bdbnn_amd.engine import Trainer, ede_inject, build_kurtosis_table
from bdbnn_amd.engine.trainer import kurtosis_targets, DIFFKURT_TARGETS
from bdbnn_amd.ops.binary_conv import _HardBinaryConvBase
from bdbnn_amd.utils import cpt_tk
from bdbnn_amd.data import dataloader_synthetic


def _args(extra=()):
    return build_parser().parse_args(["./"] + list(extra))


def test_full_flag_surface_parses():
    args = _args([
        "-a", "resnet18", "--dataset", "imagenet", "-b", "64", "--epochs", "2",
        "--ede", "--w-kurtosis", "--weight-name", "all", "--diffkurt",
        "--kurtosis-mode", "sum", "--kurtepoch", "1",
        "--imagenet_setting_step_2_ts", "-a_teacher", "resnet18",
        "--react", "--alpha", "0.5", "--beta", "100", "--temperature", "2",
        "--w-l2-reg", "--w-wr-reg", "--w-lambda-ce", "0.7",
        "--multiprocessing-distributed", "--seed", "3"])
    assert args.w_lambda_ce == 0.7
    assert args.w_l2_reg and args.w_wr_reg  # ref bug fixed: these exist
    assert args.qk_dim == 128 and not args.kd


def test_cpt_tk_schedule():
    t0, k0 = cpt_tk(0, 100)
    t_end, _ = cpt_tk(100, 100)
    assert abs(t0.item() - 1e-2) < 1e-6
    assert abs(t_end.item() - 1e1) < 1e-4
    assert k0.item() == pytest.approx(100.0)
    _, k_late = cpt_tk(90, 100)
    assert k_late.item() == 1.0


def test_ede_injection_sets_tk_on_convs():
    m = build_model(_args(["--dataset", "cifar10", "-a", "resnet20"]),
                    "resnet20", True)
    ede_inject(m, 5, 10)
    convs = [mod for mod in m.modules()
             if isinstance(mod, _HardBinaryConvBase)]
    assert all(mod.t is not None and mod.k is not None for mod in convs)


def test_kurtosis_table_excludes_first_conv():
    args = _args(["--dataset", "imagenet", "-a", "resnet18",
                  "--w-kurtosis", "--weight-name", "all"])
    m = build_model(args, "resnet18", True)
    table = build_kurtosis_table(m, args)
    assert len(table) == 19
    assert "conv1.weight" not in table
    tgts = kurtosis_targets(args, 19)
    assert tgts == [1.8] * 19
    args.diffkurt = True
    assert kurtosis_targets(args, 19) == DIFFKURT_TARGETS["imagenet"]


def _short_fit(extra, steps=6):
    args = _args(extra)
    model = build_model(args, args.arch, True)
    teacher = build_teacher(args) if args.imagenet_setting_step_2_ts else None
    trainer = Trainer(model, args, teacher=teacher,
                      device=torch.device("cpu"))
    loader = dataloader_synthetic(args.batch_size, (3, 32, 32), 10,
                                  length=args.batch_size * steps, workers=0)
    return trainer, loader


def test_train_epoch_reduces_loss():
    torch.manual_seed(0)
    trainer, loader = _short_fit(
        ["--dataset", "cifar10", "-a", "resnet20", "-b", "16",
         "--epochs", "3", "-lr", "0.05", "--w-kurtosis",
         "--weight-name", "all", "--print-freq", "100"])
    _, loss_first = trainer.train_epoch(loader, 0)
    for e in range(1, 3):
        _, loss_last = trainer.train_epoch(loader, e)
    assert loss_last < loss_first


def test_teacher_student_step_runs():
    torch.manual_seed(0)
    trainer, loader = _short_fit(
        ["--dataset", "cifar10", "-a", "resnet18", "-b", "8", "--epochs", "1",
         "-lr", "0.01", "--imagenet_setting_step_2_ts",
         "-a_teacher", "resnet18", "--print-freq", "100"], steps=2)
    assert trainer.kd_weight is not None
    top1, loss = trainer.train_epoch(loader, 0)
    assert loss == loss  # finite


def test_react_zeroes_ce_and_weight_kd():
    args = _args(["--dataset", "cifar10", "-a", "resnet18", "-b", "4",
                  "--imagenet_setting_step_2_ts", "--react",
                  "-a_teacher", "resnet18"])
    model = build_model(args, args.arch, True)
    teacher = build_teacher(args)
    trainer = Trainer(model, args, teacher=teacher, device=torch.device("cpu"))
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    total, ce, kurt, out = trainer._step_losses_ts(x, y, 0)
    with torch.no_grad():
        t_out = trainer.teacher.module(x)
    # react: total == alpha * logit-KD only (no CE, no weight-KD)
    expected = args.alpha * trainer.kd_logit(out, t_out)
    assert torch.allclose(total, expected, atol=1e-5)


def test_evaluate_mode():
    trainer, loader = _short_fit(
        ["--dataset", "cifar10", "-a", "resnet20", "-b", "8",
         "--epochs", "1"], steps=2)
    acc = trainer.validate(loader)
    assert 0.0 <= acc <= 100.0


def test_emergency_save_writes_checkpoint(tmp_path):
    args = _args(["--dataset", "cifar10", "-a", "resnet20", "-b", "4",
                  "--epochs", "1"])
    args.log_path = str(tmp_path)
    model = build_model(args, "resnet20", True)
    trainer = Trainer(model, args, device=torch.device("cpu"))
    trainer._emergency_save(epoch=2)
    import os
    assert os.path.exists(os.path.join(str(tmp_path), "checkpoint.pth.tar"))


def test_cifar100_model_path():
    args = _args(["--dataset", "cifar100", "-a", "resnet20", "-b", "4"])
    m = build_model(args, "resnet20", True)
    out = m(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, 100)
