"""modal.experimental.clustered analog: multi-rank process groups on one node.

Reference contract (14_clusters/simple_torch_cluster.py:96-118): a function
decorated ``@clustered(size=n)`` runs n replicas simultaneously; each sees
``get_cluster_info()`` → (rank, container_ips) and rank 0's IP is the
torch.distributed rendezvous master.  The single-node MI355X analog: n worker
processes, each pinned to its own GPU slice, rendezvous over 127.0.0.1 — RCCL
rides xGMI between the node's GPUs.
"""
from __future__ import annotations

import os
import socket
from dataclasses import dataclass
from typing import List


@dataclass
class ClusterInfo:
    rank: int
    container_ips: List[str]
    task_ids: List[str]


def get_cluster_info() -> ClusterInfo:
    rank = int(os.environ.get("MODAL_AMD_CLUSTER_RANK", os.environ.get("RANK", "0")))
    size = int(os.environ.get("MODAL_AMD_CLUSTER_SIZE", os.environ.get("WORLD_SIZE", "1")))
    # single node: every "container" is loopback; rendezvous must use 127.0.0.1
    ips = ["127.0.0.1"] * size
    ids = [os.environ.get("MODAL_TASK_ID", f"ta-local-{i}") for i in range(size)]
    return ClusterInfo(rank=rank, container_ips=ips, task_ids=ids)


def clustered(size: int, rdma: bool = False):
    """Decorator marking a function to run as ``size`` simultaneous ranks.

    ``fn.remote()`` launches all ranks — one dedicated worker container per
    rank with RANK / WORLD_SIZE / MASTER_ADDR=127.0.0.1 / MASTER_PORT and the
    MODAL_AMD_CLUSTER_* env injected — waits for every rank and returns
    rank 0's result (any rank's exception propagates).  ``.spawn()`` returns
    a FunctionCall over the whole gang.  Wiring lives in
    ``app.Function._submit`` + ``app.ClusterCall``."""

    def deco(fn):
        from ..app import _set_flag

        return _set_flag(fn, clustered=True, cluster_size=size)

    return deco


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p
