"""BucketedDataParallel correctness with gloo, world_size=2 (CPU CI
stand-in for RCCL on the 8-GPU node)."""

import multiprocessing as mp
import os
import pickle

import pytest
import torch
import torch.distributed as dist

from bdbnn_amd.models import cifar10 as cm


def _worker(rank, world, port, q):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from bdbnn_amd.parallel import BucketedDataParallel
        torch.manual_seed(123)  # identical init on both ranks
        model = BucketedDataParallel(cm.resnet20(), bucket_bytes=1 << 20)
        torch.manual_seed(1000 + rank)  # different data per rank
        x = torch.randn(4, 3, 32, 32)
        y = torch.randint(0, 10, (4,))
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        model.finish_gradient_sync()
        grads = {n: p.grad.clone() for n, p in model.named_parameters()
                 if p.grad is not None}
        q.put((rank, pickle.dumps({n: g.numpy() for n, g in grads.items()})))
    finally:
        dist.destroy_process_group()


def _single_rank_reference():
    torch.manual_seed(123)
    model = cm.resnet20()
    grads_sum = None
    for rank in range(2):
        torch.manual_seed(1000 + rank)
        x = torch.randn(4, 3, 32, 32)
        y = torch.randint(0, 10, (4,))
        model.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        g = {n: p.grad.clone() for n, p in model.named_parameters()}
        if grads_sum is None:
            grads_sum = g
        else:
            grads_sum = {n: grads_sum[n] + g[n] for n in g}
    return {n: v / 2 for n, v in grads_sum.items()}


@pytest.mark.timeout(300)
def test_bucketed_allreduce_matches_mean_of_ranks():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29531
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, blob = q.get(timeout=240)
        results[rank] = {n: torch.from_numpy(g.copy())
                         for n, g in pickle.loads(blob).items()}
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    # both ranks ended with identical grads
    for n in results[0]:
        assert torch.allclose(results[0][n], results[1][n], atol=1e-6), n

    # and they equal the mean of the two per-rank grads
    ref = _single_rank_reference()
    for n, g in results[0].items():
        name = n[len("module."):]
        assert torch.allclose(g, ref[name], atol=1e-5), name


def _worker_full_step(rank, world, port, q):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from bdbnn_amd.engine import Trainer
        from train import build_parser
        args = build_parser().parse_args(
            ["./", "--dataset", "cifar10", "-a", "resnet20", "-b", "8",
             "--epochs", "1", "-lr", "0.05", "--w-kurtosis",
             "--weight-name", "all"])
        torch.manual_seed(99)
        from bdbnn_amd.models import cifar10 as cm2
        model = cm2.resnet20()
        trainer = Trainer(model, args, device=torch.device("cpu"),
                          world_size=world, rank=rank)
        torch.manual_seed(500 + rank)
        x = torch.randn(4, 3, 32, 32)
        y = torch.randint(0, 10, (4,))
        total, ce, kurt, out = trainer._step_losses(x, y, 0)
        trainer.optimizer.zero_grad(set_to_none=True)
        total.backward()
        trainer.model.finish_gradient_sync()
        trainer.optimizer.step()
        with torch.no_grad():
            h = sum(float(p.double().sum()) for p in
                    trainer.model.parameters())
        q.put((rank, h))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_rank_full_step_params_identical():
    """After one full step (fwd + kurtosis + bwd + all-reduce + fused
    optimizer), parameters must be bitwise-consistent across ranks."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_full_step, args=(r, 2, 29537, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        rank, h = q.get(timeout=240)
        res[rank] = h
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert res[0] == res[1]
