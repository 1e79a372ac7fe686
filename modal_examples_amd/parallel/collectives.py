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


# message-size-aware all-reduce algorithm selection:
# xGMI is 7 point-to-point links; a ring all-reduce moves 2·(w-1)/w of the
# data over ONE link in 2(w-1) latency-bound steps.  For small latency-
# critical TP messages a ONE-SHOT all-gather + local reduce (each GPU pulls
# every peer's shard concurrently over its 7 links, one step) wins; for
# MB-scale+ payloads the ring's lower traffic wins.  128 KiB is the
# crossover used by one-shot implementations at 8-GPU scale.
ONESHOT_MAX_BYTES = 128 * 1024


def all_reduce_smart(t: torch.Tensor, threshold: Optional[int] = None):
    """All-reduce with algorithm selection by message size.

    <= threshold bytes: one-shot (all_gather into w buffers — every link
    active in one step — then a local sum).  > threshold: library ring.
    Correct on gloo/CPU too (tested hermetically at world 4)."""
    if not dist.is_initialized():
        return t
    thr = ONESHOT_MAX_BYTES if threshold is None else threshold
    nbytes = t.numel() * t.element_size()
    if nbytes <= thr:
        w = dist.get_world_size()
        gathered = [torch.empty_like(t) for _ in range(w)]
        dist.all_gather(gathered, t)
        acc = gathered[0]
        for g in gathered[1:]:
            acc = acc + g
        t.copy_(acc)
        return t
    dist.all_reduce(t)
    return t
