#!/usr/bin/env python3
"""BD-BNN training CLI — flag-compatible with the reference's train.py
(ref:train.py:64-171; full flag inventory in SURVEY.md section 2.1).

MI355X-native underneath: one process per GPU over RCCL/xGMI (launch
with torchrun / torch.distributed.run, or --multiprocessing-distributed
to spawn locally), binary convs on the bit-packed XNOR kernel, fused
kurtosis/KD/optimizer kernels.

Latent reference bugs fixed but kept flag-compatible:
* --w-l2-reg / --w-wr-reg exist (ref reads undefined attrs, train.py:480)
* --w-lambda-ce defaults to 1.0 (ref crashes non-react TS runs)
* distributed sampler actually shards the dataset per rank
"""

import argparse
import logging
import os
import random
import sys
import time

import torch
import torch.nn as nn

import bdbnn_amd
from bdbnn_amd import models as _models
from bdbnn_amd.data import (dataloader_cifar10, dataloader_cifar100,
                            dataloader_imagenet)
from bdbnn_amd.engine import Trainer
from bdbnn_amd.parallel import init_distributed

cifar_models = _models.cifar10
imagenet_models = _models.imagenet


def model_names(ns):
    return sorted(n for n in ns.__dict__
                  if not n.startswith("_") and callable(ns.__dict__[n]))


def build_parser():
    p = argparse.ArgumentParser(description="BD-BNN (MI355X-native) training")
    p.add_argument("data", metavar="DIR", nargs="?", default="./",
                   help="path to dataset root")
    p.add_argument("-a", "--arch", default="resnet18")
    p.add_argument("-j", "--workers", default=4, type=int)
    p.add_argument("--epochs", default=90, type=int)
    p.add_argument("--start-epoch", default=0, type=int)
    p.add_argument("-b", "--batch-size", default=256, type=int,
                   help="total batch size across the node; divided per rank")
    p.add_argument("-lr", "--learning-rate", default=0.1, type=float,
                   dest="lr")
    p.add_argument("--momentum", default=0.9, type=float)
    p.add_argument("-wd", "--weight-decay", default=1e-4, type=float,
                   dest="weight_decay")
    p.add_argument("-p", "--print-freq", default=10, type=int)
    p.add_argument("--resume", default="", type=str)
    p.add_argument("-e", "--evaluate", action="store_true")
    p.add_argument("--pretrained", action="store_true")
    p.add_argument("--world-size", default=1, type=int)
    p.add_argument("--rank", default=0, type=int)
    p.add_argument("--dist-url", default="tcp://127.0.0.1:23456", type=str)
    p.add_argument("--master-addr", default="127.0.0.1", type=str)
    p.add_argument("--dist-backend", default="nccl", type=str)
    p.add_argument("--seed", default=None, type=int)
    p.add_argument("--gpu", default=None, type=int)
    p.add_argument("--multiprocessing-distributed", action="store_true")
    p.add_argument("--log_path", default="log", type=str)
    p.add_argument("--custom_resnet", action="store_true")
    p.add_argument("--reset_resume", action="store_true")
    p.add_argument("--ede", action="store_true")
    p.add_argument("--w-kurtosis-target", default=1.8, type=float)
    p.add_argument("--w-lambda-kurtosis", default=1.0, type=float)
    p.add_argument("--w-kurtosis", action="store_true")
    p.add_argument("--weight-name", nargs="+", default=None)
    p.add_argument("--remove-weight-name", nargs="+", default=None)
    p.add_argument("--kurtosis-mode", default="avg",
                   choices=("avg", "sum", "max"))
    p.add_argument("--diffkurt", action="store_true")
    p.add_argument("--kurtepoch", default=0, type=int)
    p.add_argument("--twoblock", action="store_true")  # parity: unused upstream
    p.add_argument("--dataset", default="cifar10",
                   choices=("cifar10", "cifar100", "imagenet"))
    p.add_argument("--imagenet_setting", action="store_true")
    p.add_argument("--imagenet_setting_step_1", action="store_true")
    p.add_argument("--imagenet_setting_step_2", action="store_true")
    p.add_argument("--imagenet_setting_step_2_ts", action="store_true",
                   help="enable the teacher-student path")
    p.add_argument("-a_teacher", "--arch_teacher", default="resnet18")
    p.add_argument("--custom_resnet_teacher", action="store_true")
    p.add_argument("--resume_teacher", default="", type=str)
    p.add_argument("--kd", action="store_true")  # parity: unused upstream
    p.add_argument("--react", action="store_true")
    p.add_argument("--alpha", default=0.9, type=float)
    p.add_argument("--temperature", default=4, type=float)
    p.add_argument("--beta", default=200, type=float)
    p.add_argument("--qk_dim", default=128, type=int)  # parity: unused upstream
    # fixed latent-bug flags (ref reads these without defining them)
    p.add_argument("--w-l2-reg", action="store_true", dest="w_l2_reg")
    p.add_argument("--w-wr-reg", action="store_true", dest="w_wr_reg")
    p.add_argument("--w-lambda-ce", default=1.0, type=float,
                   dest="w_lambda_ce")
    # our additions
    p.add_argument("--amp", action="store_true",
                   help="bf16 autocast for the dense (stem/head/backward) path")
    p.add_argument("--synthetic-data", action="store_true",
                   help="synthetic dataset of the selected shape (offline)")
    p.add_argument("--synthetic-train-len", default=100000, type=int,
                   help="length of the synthetic imagenet train split "
                        "(short epochs for smoke/parity runs)")
    p.add_argument("--auto-resume", action="store_true",
                   help="resume from <log_path>/checkpoint.pth.tar if present "
                        "(crash recovery; the reference requires manual --resume)")
    return p


def build_model(args, arch, custom):
    num_classes = {"cifar10": 10, "cifar100": 100, "imagenet": 1000}[args.dataset]
    if args.dataset == "imagenet":
        ctor = imagenet_models.__dict__.get(arch)
        if ctor is None:
            raise SystemExit(f"unknown imagenet arch {arch!r}; "
                             f"have {model_names(imagenet_models)}")
        return ctor(args.pretrained)
    ctor = cifar_models.__dict__.get(arch)
    if ctor is None:
        raise SystemExit(f"unknown cifar arch {arch!r}; "
                         f"have {model_names(cifar_models)}")
    return ctor(num_classes=num_classes)


def build_teacher(args):
    """fp32 teacher (frozen; co-resident with the student in HBM)."""
    if args.dataset == "imagenet":
        name = args.arch_teacher
        if not name.endswith("_real"):
            name = name + "_real"
        ctor = imagenet_models.__dict__.get(name) or imagenet_models.__dict__.get(args.arch_teacher)
    else:
        name = args.arch_teacher
        if not name.endswith("_real"):
            name = name + "_real"
        ctor = cifar_models.__dict__.get(name)
        if ctor is not None:
            num_classes = {"cifar10": 10, "cifar100": 100}[args.dataset]
            return ctor(num_classes=num_classes)
    if ctor is None:
        raise SystemExit(f"unknown teacher arch {args.arch_teacher!r}")
    return ctor(True)


def make_loaders(args, distributed):
    per_rank_bs = args.batch_size
    if distributed:
        import torch.distributed as dist
        per_rank_bs = max(1, args.batch_size // dist.get_world_size())
    kw = dict(batch_size=per_rank_bs, data_path=args.data,
              workers=args.workers, synthetic=args.synthetic_data)
    if args.dataset == "imagenet":
        kw["synthetic_len"] = args.synthetic_train_len
    fn = {"cifar10": dataloader_cifar10, "cifar100": dataloader_cifar100,
          "imagenet": dataloader_imagenet}[args.dataset]
    train_loader = fn("train", distributed=distributed, **kw)
    val_loader = fn("val", distributed=False, **kw)
    return train_loader, val_loader


def setup_logging(args, rank):
    os.makedirs(args.log_path, exist_ok=True)
    handlers = [logging.StreamHandler(sys.stdout)]
    if rank == 0:
        handlers.append(logging.FileHandler(
            os.path.join(args.log_path, "log.txt")))
    logging.basicConfig(level=logging.INFO, handlers=handlers,
                        format="%(asctime)s %(name)s: %(message)s", force=True)


def main(argv=None):
    args = build_parser().parse_args(argv)

    if args.seed is not None:
        random.seed(args.seed)
        torch.manual_seed(args.seed)

    # log dir: <log_path>/<kurtosis_target>/<timestamp> (ref:train.py:189-190)
    stamp = time.strftime("%Y-%m-%d_%H-%M-%S")
    args.log_path = os.path.join(args.log_path,
                                 str(args.w_kurtosis_target), stamp)

    # honor the reference's rendezvous flags when torchrun env is absent
    if "MASTER_ADDR" not in os.environ:
        if args.dist_url.startswith("tcp://"):
            host, _, port = args.dist_url[len("tcp://"):].partition(":")
            os.environ["MASTER_ADDR"] = host or args.master_addr
            if port:
                os.environ.setdefault("MASTER_PORT", port)
        elif args.master_addr:
            os.environ["MASTER_ADDR"] = args.master_addr

    if args.multiprocessing_distributed and "RANK" not in os.environ:
        # spawn one process per GPU; multi-node: --world-size nodes with
        # --rank node_rank each (the reference's rank math is broken for
        # this case, SURVEY.md 2.1 — here rank = node_rank*ngpus + gpu)
        ngpus = max(torch.cuda.device_count(), 1)
        if ngpus > 1 or args.world_size > 1:
            import torch.multiprocessing as mp
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29513")
            mp.spawn(_spawned,
                     nprocs=ngpus,
                     args=(ngpus, args.world_size * ngpus,
                           args.rank * ngpus, argv))
            return None
    return _run(args)


def _spawned(local_rank, ngpus, world_size, rank_base, argv):
    os.environ["RANK"] = str(rank_base + local_rank)
    os.environ["LOCAL_RANK"] = str(local_rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    args = build_parser().parse_args(argv)
    stamp = time.strftime("%Y-%m-%d_%H-%M-%S")
    args.log_path = os.path.join(args.log_path,
                                 str(args.w_kurtosis_target), stamp)
    _run(args)


def _run(args):
    rank, local_rank, world_size = init_distributed(
        backend=None if args.dist_backend == "nccl" else args.dist_backend)
    if world_size > 1:
        os.environ.setdefault("MIOPEN_USER_DB_PATH",
                              f"/tmp/miopen-rank{local_rank}")
    setup_logging(args, rank)
    log = logging.getLogger("bdbnn")
    log.info(f"bdbnn_amd {bdbnn_amd.__version__} rank {rank}/{world_size}")

    model = build_model(args, args.arch, args.custom_resnet)
    teacher = build_teacher(args) if args.imagenet_setting_step_2_ts else None

    if torch.cuda.is_available():
        dev_idx = (args.gpu if args.gpu is not None
                   else local_rank % max(torch.cuda.device_count(), 1))
        device = torch.device(f"cuda:{dev_idx}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    trainer = Trainer(model, args, teacher=teacher, device=device,
                      world_size=world_size, rank=rank)

    if args.resume and os.path.isfile(args.resume):
        trainer.resume(args.resume, reset_resume=args.reset_resume)
    elif args.auto_resume:
        auto_ckpt = os.path.join(os.path.dirname(os.path.dirname(args.log_path)),
                                 "checkpoint.pth.tar")
        if os.path.isfile(auto_ckpt):
            log.info(f"auto-resume from {auto_ckpt}")
            trainer.resume(auto_ckpt)
    if teacher is not None:
        if args.resume_teacher:
            if not os.path.isfile(args.resume_teacher):
                raise SystemExit(
                    f"--resume_teacher {args.resume_teacher!r}: no such "
                    "file (a TS run distilling from a random-init teacher "
                    "is almost certainly not what you want)")
            from bdbnn_amd.engine.checkpoint import load_state
            load_state(args.resume_teacher, trainer.teacher,
                       map_location=str(device))
        else:
            log.warning(
                "teacher-student run without --resume_teacher: the "
                "teacher is RANDOM-INIT (no pretrained weights in this "
                "offline image) — KD will distill noise")

    train_loader, val_loader = make_loaders(args, world_size > 1)

    if args.evaluate:
        acc = trainer.validate(val_loader)
        log.info(f"test acc: {acc}")
        return acc

    best = trainer.fit(train_loader, val_loader)
    log.info(f"done; best acc1 {best:.3f}; log {args.log_path}")
    return best


if __name__ == "__main__":
    main()
