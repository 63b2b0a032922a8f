"""Process-group bring-up: one process per GPU over RCCL/xGMI.

Replaces the reference's Spark local[4] + Kryo tensor serialization
(Java:316-322): ranks come from torchrun-style env vars, tensors move over
RCCL collectives on xGMI instead of being serialized driver->executor.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str = "auto", timeout_s: int = 600):
    """Initialize from RANK/WORLD_SIZE/LOCAL_RANK/MASTER_* env.

    Returns (rank, world_size, local_rank, device). Single-process friendly:
    with no env vars set, returns (0, 1, 0, best-device) without creating a
    process group (the reference's "runs without a cluster" property,
    SURVEY.md §4 item 4).
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    cuda = torch.cuda.is_available()
    device = torch.device("cuda", local_rank % max(1, torch.cuda.device_count())) \
        if cuda else torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        backend = os.environ.get("GDLJ_BACKEND", backend)
        if backend == "auto":
            backend = "nccl" if cuda else "gloo"
        if cuda and backend == "nccl":
            torch.cuda.set_device(device)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    elif cuda:
        torch.cuda.set_device(device)
    return rank, world, local_rank, device


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def is_main() -> bool:
    return get_rank() == 0


def barrier():
    if dist.is_initialized():
        dist.barrier()
