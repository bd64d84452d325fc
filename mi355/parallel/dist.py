"""Process-group bootstrap for one-process-per-GPU data parallelism.

The reference delegates this to Ray Train setting MASTER_ADDR/RANK/WORLD_SIZE
on its worker actors and calling torch.distributed.init_process_group
(flan-t5-batch-inference.py:96-98). Here the same env-var
rendezvous is used; on ROCm the "nccl" backend IS RCCL over xGMI.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_s: int = 300) -> int:
    """Initialise torch.distributed from env vars; returns the rank.

    No-op (returns 0) when WORLD_SIZE is absent/1 and not already initialised.
    """
    if dist.is_initialized():
        return dist.get_rank()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend == "nccl":
        # LOCAL_RANK modulo the VISIBLE device count: trainer workers are
        # pinned via HIP_VISIBLE_DEVICES (each sees ONE device, index 0,
        # while LOCAL_RANK stays the global rank), torchrun ranks are
        # unpinned (count == world). Without the modulo every pinned rank>0
        # died on an invalid ordinal before rendezvous.
        torch.cuda.set_device(local_rank() % max(1, torch.cuda.device_count()))
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_s))
    return dist.get_rank()


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def is_main_process() -> bool:
    return get_rank() == 0


def barrier():
    if is_distributed():
        dist.barrier()
