"""Process-group helpers for one-process-per-GPU RCCL/xGMI training."""
from __future__ import annotations

import functools
import os
from typing import Optional

import torch
import torch.distributed as dist


def init_distributed_from_env(backend: Optional[str] = None) -> int:
    """Initialize torch.distributed from torchrun-style env vars (RANK/LOCAL_RANK/
    WORLD_SIZE/MASTER_ADDR/MASTER_PORT). Returns local rank. No-op when WORLD_SIZE
    is 1 or unset. Backend defaults to "nccl" (= RCCL on ROCm) when a GPU is
    visible, "gloo" otherwise."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return local_rank


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def is_main_process() -> bool:
    return get_rank() == 0


def rank_zero_only(fn):
    """Decorator: run only on rank 0 (parity with Lightning's @rank_zero_only)."""

    @functools.wraps(fn)
    def wrapped(*args, **kwargs):
        if is_main_process():
            return fn(*args, **kwargs)
        return None

    return wrapped


def split_dataset_by_node(dataset, rank: Optional[int] = None, world_size: Optional[int] = None):
    """Shard a (streaming) 🤗 dataset across ranks (parity with the reference's only
    explicit torch.distributed call site, data/text/c4.py:57-79)."""
    from datasets.distributed import split_dataset_by_node as _split

    if rank is None:
        rank = get_rank()
    if world_size is None:
        world_size = get_world_size()
    if world_size == 1:
        return dataset
    return _split(dataset, rank=rank, world_size=world_size)
