"""Process-group bring-up: one process per GPU, RCCL over xGMI.

Replaces the reference's L2 (ref:train.py:196-249): ``torch.distributed``
with backend "nccl" IS RCCL on ROCm; on CPU-only machines (CI) we use
gloo.  Rendezvous reads the standard torchrun env vars
(RANK/LOCAL_RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT).
"""

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend=None, timeout_s=300):
    """Initialise the process group from torchrun-style env vars.

    Returns (rank, local_rank, world_size).  Safe to call when
    WORLD_SIZE is absent/1: returns (0, 0, 1) without creating a group.
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    if world_size <= 1:
        return 0, local_rank, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s))
    return rank, local_rank, world_size


def get_rank():
    return dist.get_rank() if dist.is_available() and dist.is_initialized() else 0


def get_world_size():
    return dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1


def barrier():
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
