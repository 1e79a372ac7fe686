"""RCCL-over-xGMI collective helpers.

Design per SURVEY.md §5.8: torch.distributed's "nccl" backend IS RCCL on ROCm;
the 8-GPU MI355X node is fully connected with 7 point-to-point xGMI links per
GPU (~153 GB/s each), so ring collectives are per-link bound — gradient
buckets are sized for link bandwidth (50 MB default) and overlapped with
backward on the compute stream; GB-scale weight broadcasts just ride the ring.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None) -> tuple:
    """Initialize from torchrun env (RANK/WORLD_SIZE/LOCAL_RANK). Returns
    (rank, world, local_rank); no-op (0,1,0) when not launched distributed."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1, 0
    rank = int(os.environ.get("RANK", "0"))
    local = int(os.environ.get("LOCAL_RANK", "0"))
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world)
        if backend == "nccl":
            torch.cuda.set_device(local)
    return rank, world, local


def broadcast_module(module: torch.nn.Module, src: int = 0):
    """Weight distribution for .map fan-out (SURVEY.md §2.3 DP row)."""
    if not dist.is_initialized():
        return
    for p in module.state_dict().values():
        if isinstance(p, torch.Tensor):
            dist.broadcast(p, src=src)


def all_reduce_mean(t: torch.Tensor):
    if dist.is_initialized():
        dist.all_reduce(t)
        t /= dist.get_world_size()
    return t
