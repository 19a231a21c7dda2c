#!/usr/bin/env python3
"""Flagship benchmark: BD-BNN ResNet-18 training step on ImageNet-shaped
synthetic data (BASELINE.json metric: train images/sec, node aggregate).

    python bench.py --gpus N --steps K --warmup W
    # N>1: launched by the driver as torch.distributed.run, one rank/GPU
    # over RCCL; reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env.

The timed step is the full training step of the BD-BNN method on the
named config: binary-conv forward (XNOR kernel), CE + kurtosis losses,
backward (dense MFMA convs + fused mask/kurtosis kernels), bucketed
gradient all-reduce, fused optimizer step.  --ts adds the fp32-teacher
KD path (BASELINE config 3).  Rank 0 prints ONE JSON line.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=2048,
                   help="per-GPU batch size")
    p.add_argument("--arch", default="resnet18")
    p.add_argument("--ts", action="store_true",
                   help="teacher-student KD path (BASELINE config 3)")
    p.add_argument("--infer", action="store_true",
                   help="binary inference path: packed weights + hipGraph "
                        "(BASELINE config 5, throughput-only)")
    p.add_argument("--no-kurt", action="store_true")
    p.add_argument("--diffkurt", action="store_true",
                   help="per-layer kurtosis target lists (BASELINE config 4)")
    p.add_argument("--image", type=int, default=224)
    return p.parse_args()


def main():
    args = parse_args()
    import bdbnn_amd  # noqa: F401
    from bdbnn_amd.engine import Trainer
    from bdbnn_amd.models import imagenet as im
    from bdbnn_amd.parallel import init_distributed

    rank, local_rank, world_size = init_distributed()
    if world_size > 1:
        # per-rank MIOpen find db: concurrent ranks otherwise serialize on
        # the shared user-db file lock during warmup
        os.environ.setdefault("MIOPEN_USER_DB_PATH",
                              f"/tmp/miopen-rank{local_rank}")
    use_cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}") if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    torch.manual_seed(1234 + rank)

    if not use_cuda and args.batch_size >= 512:
        # CPU-only machine (no-flag contract: must finish in minutes)
        args.batch_size = 16
        args.image = min(args.image, 64)

    class TArgs:  # minimal trainer config (mirrors train.py flags)
        arch = args.arch
        dataset = "imagenet"
        lr = 1e-3
        momentum = 0.9
        weight_decay = 1e-4
        epochs = 90
        w_kurtosis = not args.no_kurt
        weight_name = ["all"]
        remove_weight_name = None
        w_kurtosis_target = 1.8
        w_lambda_kurtosis = 1.0
        kurtosis_mode = "avg"
        diffkurt = args.diffkurt
        kurtepoch = 0
        react = False
        alpha = 0.9
        beta = 200.0
        w_lambda_ce = 1.0
        amp = use_cuda
        print_freq = 10**9
        ede = False
        start_epoch = 0

    model = im.__dict__[args.arch](False)

    if args.infer:
        from bdbnn_amd.engine import PackedInference
        assert use_cuda, "--infer needs a GPU"
        eng = PackedInference(model).capture(
            (args.batch_size, 3, args.image, args.image))
        x = torch.randn(args.batch_size, 3, args.image, args.image,
                        device=device)
        for _ in range(args.warmup):
            eng(x)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            eng(x)
        torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        print(json.dumps({
            "metric": "infer_images_per_sec",
            "value": round(args.batch_size * args.steps / elapsed, 2),
            "unit": "images/s", "n_gpus": 1, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "bf16", "data": "synthetic",
            "config": {"model": f"{args.arch}_bdbnn_infer",
                       "global_batch": args.batch_size,
                       "image": f"3x{args.image}x{args.image}",
                       "parallelism": "dp1", "hipgraph": True}}))
        return

    teacher = im.__dict__[args.arch + "_real"](False) if args.ts else None
    trainer = Trainer(model, TArgs, teacher=teacher, device=device,
                      world_size=world_size, rank=rank)

    # synthetic ImageNet-shaped data, random-init weights (no datasets in
    # this offline image); a few fixed batches resident on device
    n_batches = 4
    batches = []
    for i in range(n_batches):
        x = torch.randn(args.batch_size, 3, args.image, args.image,
                        device=device)
        if use_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        y = torch.randint(0, 1000, (args.batch_size,), device=device)
        batches.append((x, y))

    trainer.model.train()

    def step(i):
        x, y = batches[i % n_batches]
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=TArgs.amp):
            total, ce, kurt, out = (
                trainer._step_losses_ts(x, y, 0) if args.ts
                else trainer._step_losses(x, y, 0))
        trainer.optimizer.zero_grad(set_to_none=True)
        total.backward()
        trainer.model.finish_gradient_sync()
        trainer.optimizer.step()
        return total

    for i in range(args.warmup):
        step(i)

    if dist.is_available() and dist.is_initialized():
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist.is_available() and dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    n = world_size if world_size > 1 else args.gpus
    images_per_sec = args.batch_size * n * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": "train_images_per_sec",
            "value": round(images_per_sec, 2),
            "unit": "images/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"{args.arch}_bdbnn" + ("_ts" if args.ts else ""),
                "global_batch": args.batch_size * n,
                "image": f"3x{args.image}x{args.image}",
                "kurtosis": not args.no_kurt,
                "parallelism": f"dp{n}",
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
