# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/14_clusters/simple_torch_cluster.py"]
# ---
# # Multi-GPU process groups (the clusters example, single-node analog)
#
# `@modal.experimental.clustered(size=n)` runs n simultaneous ranks;
# `get_cluster_info()` provides rank + peer IPs (loopback on one node), and
# torch.distributed rendezvous over 127.0.0.1 — RCCL over xGMI between GPUs,
# gloo on CPU.  The payload: rank 0 broadcasts a tensor, every rank verifies.

import os
import subprocess
import sys

import modal_examples_amd as modal

app = modal.App("example-torch-cluster")

WORLD = int(os.environ.get("CLUSTER_WORLD", "2"))

WORKER = r"""
import os, torch, torch.distributed as dist
from modal_examples_amd.parallel.cluster import get_cluster_info

info = get_cluster_info()
rank, world = info.rank, len(info.container_ips)
backend = "nccl" if torch.cuda.is_available() else "gloo"
dist.init_process_group(backend, rank=rank, world_size=world,
                        init_method=f"tcp://127.0.0.1:{os.environ['CLUSTER_PORT']}")
dev = f"cuda:{rank}" if backend == "nccl" else "cpu"
if backend == "nccl":
    torch.cuda.set_device(rank)
t = torch.full((4,), float(rank), device=dev)
if rank == 0:
    t.fill_(42.0)
dist.broadcast(t, src=0)
assert t.eq(42).all(), t
dist.barrier()
print(f"rank {rank}/{world} on {dev}: broadcast ok -> {t.tolist()}")
dist.destroy_process_group()
"""


@app.function(gpu=f"mi355x:{WORLD}", timeout=300)
def run_cluster() -> int:
    """Launches one process per GPU (torchrun-style) inside the allocation."""
    from modal_examples_amd.parallel.cluster import free_port

    port = free_port()
    procs = []
    for rank in range(WORLD):
        env = dict(os.environ)
        env.update({
            "MODAL_AMD_CLUSTER_RANK": str(rank),
            "MODAL_AMD_CLUSTER_SIZE": str(WORLD),
            "CLUSTER_PORT": str(port),
            "MASTER_ADDR": "127.0.0.1",
        })
        procs.append(subprocess.Popen([sys.executable, "-c", WORKER], env=env))
    rcs = [p.wait(timeout=240) for p in procs]
    assert all(rc == 0 for rc in rcs), rcs
    return WORLD


@app.local_entrypoint()
def main():
    n = run_cluster.remote()
    print(f"cluster of {n} ranks completed the broadcast")
