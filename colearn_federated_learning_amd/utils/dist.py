"""Process-group helpers: one rank per GPU, RCCL over xGMI (gloo on CPU).

``torch.distributed`` backend "nccl" IS RCCL on ROCm.  Launch shape is
torchrun-style: RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT
in the environment.
"""

from __future__ import annotations

import datetime
import os
from typing import Tuple

import torch
import torch.distributed as dist


def rank_world() -> Tuple[int, int, int]:
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    return rank, world, local_rank


def local_device() -> torch.device:
    _, _, local_rank = rank_world()
    if torch.cuda.is_available():
        return torch.device("cuda", local_rank % torch.cuda.device_count())
    return torch.device("cpu")


def init_distributed(backend: str = "auto") -> Tuple[int, int, torch.device]:
    """Initialize the default process group if WORLD_SIZE > 1.

    Returns (rank, world_size, device).  backend "auto" picks RCCL when a
    GPU is visible, gloo otherwise.
    """
    rank, world, local_rank = rank_world()
    device = local_device()
    if world > 1 and not dist.is_initialized():
        if backend == "auto":
            backend = "nccl" if device.type == "cuda" else "gloo"
        if device.type == "cuda":
            torch.cuda.set_device(device)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29711")
        # COLEARN_INIT_FILE: file-store rendezvous (no TCP listener) —
        # used by the multi-process test harness, where back-to-back
        # TCPStore setups occasionally raced on teardown
        init_file = os.environ.get("COLEARN_INIT_FILE")
        kwargs = dict(backend=backend, rank=rank, world_size=world,
                      timeout=datetime.timedelta(seconds=300))
        if init_file:
            kwargs["init_method"] = f"file://{init_file}"
        dist.init_process_group(**kwargs)
    elif device.type == "cuda":
        torch.cuda.set_device(device)
    return rank, world, device
