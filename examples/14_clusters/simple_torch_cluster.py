# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/14_clusters/simple_torch_cluster.py"]
# ---
# # Multi-GPU process groups (the clusters example, single-node analog)
#
# `@modal.experimental.clustered(size=n)` makes `fn.remote()` launch n
# simultaneous rank containers — the RUNTIME does the fan-out (reference
# contract: 14_clusters/simple_torch_cluster.py:96-130): each rank worker gets
# RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT injected, `get_cluster_info()`
# reports rank + peer IPs (loopback on one node), and the caller receives
# rank 0's return value once every rank finished.  torch.distributed
# rendezvous over 127.0.0.1 — RCCL over xGMI between pinned GPUs, gloo on CPU.

import modal_examples_amd as modal

app = modal.App("example-torch-cluster")

WORLD = 2


def _gpu_request():
    try:
        import torch

        return f"mi355x:1" if torch.cuda.device_count() >= WORLD else None
    except Exception:
        return None


@app.function(gpu=_gpu_request(), timeout=300)
@modal.experimental.clustered(size=WORLD)
def all_ranks_broadcast() -> str:
    """Runs in EVERY rank container; rank 0's return value is the caller's."""
    import torch
    import torch.distributed as dist

    from modal_examples_amd.parallel.cluster import get_cluster_info

    import os

    info = get_cluster_info()
    rank, world = info.rank, len(info.container_ips)
    # RCCL only when this rank has its OWN pinned GPU (the runtime sets
    # HIP_VISIBLE_DEVICES per worker); on a box with fewer GPUs than ranks
    # the gang runs gloo — two ranks on one device is an RCCL error
    use_gpu = torch.cuda.is_available() and bool(os.environ.get("HIP_VISIBLE_DEVICES"))
    backend = "nccl" if use_gpu else "gloo"  # "nccl" IS RCCL on ROCm
    dist.init_process_group(backend, init_method="env://")
    dev = "cuda:0" if use_gpu else "cpu"  # each rank sees only its pinned GPU
    t = torch.full((4,), float(rank), device=dev)
    if rank == 0:
        t.fill_(42.0)
    dist.broadcast(t, src=0)
    assert t.eq(42).all(), t
    dist.barrier()
    print(f"rank {rank}/{world} [{backend}] broadcast ok -> {t.tolist()}")
    dist.destroy_process_group()
    return f"rank0 of {world} [{backend}]"


@app.local_entrypoint()
def main():
    out = all_ranks_broadcast.remote()
    print(f"cluster done: {out}")
