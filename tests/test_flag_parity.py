"""CLI parity: every flag of the reference's argparse surface
(ref:train.py:64-171) must be accepted."""

import pytest

from train import build_parser

# (flag, sample value or None for store_true) — the complete reference list
REFERENCE_FLAGS = [
    ("-a", "resnet18"), ("--arch", "resnet18"),
    ("-j", "2"), ("--workers", "2"),
    ("--epochs", "3"), ("--start-epoch", "1"),
    ("-b", "32"), ("--batch-size", "32"),
    ("-lr", "0.01"), ("--learning-rate", "0.01"),
    ("--momentum", "0.8"),
    ("-wd", "1e-5"), ("--weight-decay", "1e-5"),
    ("-p", "5"), ("--print-freq", "5"),
    ("--resume", "x.pth.tar"),
    ("-e", None), ("--evaluate", None),
    ("--pretrained", None),
    ("--world-size", "2"), ("--rank", "1"),
    ("--dist-url", "tcp://1.2.3.4:1234"),
    ("--master-addr", "1.2.3.4"),
    ("--dist-backend", "gloo"),
    ("--seed", "3"), ("--gpu", "0"),
    ("--multiprocessing-distributed", None),
    ("--log_path", "mylog"),
    ("--custom_resnet", None), ("--reset_resume", None),
    ("--ede", None),
    ("--w-kurtosis-target", "1.4"),
    ("--w-lambda-kurtosis", "0.5"),
    ("--w-kurtosis", None),
    ("--weight-name", "all"),
    ("--remove-weight-name", "downsample"),
    ("--kurtosis-mode", "sum"),
    ("--diffkurt", None), ("--kurtepoch", "2"),
    ("--twoblock", None),
    ("--dataset", "imagenet"),
    ("--imagenet_setting", None),
    ("--imagenet_setting_step_1", None),
    ("--imagenet_setting_step_2", None),
    ("--imagenet_setting_step_2_ts", None),
    ("-a_teacher", "resnet18"), ("--arch_teacher", "resnet18"),
    ("--custom_resnet_teacher", None),
    ("--resume_teacher", "t.pth.tar"),
    ("--kd", None), ("--react", None),
    ("--alpha", "0.5"), ("--temperature", "2"), ("--beta", "100"),
    ("--qk_dim", "64"),
]


@pytest.mark.parametrize("flag,value", REFERENCE_FLAGS,
                         ids=[f for f, _ in REFERENCE_FLAGS])
def test_reference_flag_accepted(flag, value):
    argv = ["./", flag] + ([value] if value is not None else [])
    args = build_parser().parse_args(argv)
    assert args is not None


def test_reference_defaults_match():
    """Defaults the reference documents."""
    a = build_parser().parse_args(["./"])
    assert a.arch == "resnet18"
    assert a.workers == 4
    assert a.epochs == 90
    assert a.batch_size == 256
    assert a.lr == 0.1
    assert a.momentum == 0.9
    assert a.weight_decay == 1e-4
    assert a.print_freq == 10
    assert a.w_kurtosis_target == 1.8
    assert a.w_lambda_kurtosis == 1.0
    assert a.kurtosis_mode == "avg"
    assert a.dataset == "cifar10"
    assert a.alpha == 0.9
    assert a.beta == 200
    assert a.temperature == 4
    assert a.qk_dim == 128
    assert a.dist_backend == "nccl"
    assert a.log_path == "log"
