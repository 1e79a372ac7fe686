#!/usr/bin/env python3
"""Flagship benchmark: SDXL 1024px txt2img images/sec, whole node.

Contract (driver-facing): `python bench.py --gpus N --steps K --warmup W`
runs the SDXL-class pipeline (2.6B-param UNet, bf16, synthetic prompts,
random-init weights, 4 denoise steps + VAE decode per image batch) on N GPUs
of one node — rank-per-GPU data parallelism over RCCL.  One "step" = one
generate() call of `--batch` images on every rank.  Rank 0 prints ONE JSON
line with the whole-job aggregate images/sec.

Reference anchor: SD3.5-Large-Turbo/SDXL-class 1024px ≈ 0.5-1
image/s on one H100 (text_to_image.py:11-13); vs_baseline divides by
1.0 img/s × n_gpus (the reference's BEST case, scaled to node size).
"""
from __future__ import annotations

import os as _os
_os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
import argparse
import json
import os
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3, help="timed generate() calls")
    ap.add_argument("--warmup", type=int, default=1)
    # batch 16 is the measured throughput knee with the r2 hand-conv kernels
    # (scripts/sweep_batch.py: 13.3 img/s @B4 -> 16.2 @B16, 16.3 @B24,
    # declining @B32) and the top of the reference's own stated batch range
    # (text_to_image.py:13 "batch 1-16")
    ap.add_argument("--batch", type=int, default=16, help="images per rank per step")
    ap.add_argument("--denoise-steps", type=int, default=4)
    ap.add_argument("--latent", type=int, default=128, help="128 → 1024px")
    ap.add_argument("--small", action="store_true", help="test-size model")
    ap.add_argument("--no-graph", action="store_true")
    args = ap.parse_args()

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist_on = world > 1
    if dist_on:
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world)
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
    device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"

    from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
    from modal_examples_amd.models.sdxl.unet import UNetConfig

    cfg = UNetConfig.small() if args.small else UNetConfig.sdxl()
    t_init0 = time.perf_counter()
    pipe = SDXLPipeline(cfg, device=device, latent_size=args.latent,
                        use_graph=not args.no_graph, seed=1234 + rank)
    prompts = [f"benchmark prompt {rank}-{i}" for i in range(args.batch)]

    def one_step():
        img = pipe.generate(prompts, steps=args.denoise_steps, guidance=0.0)
        return img

    # warmup (includes graph capture); first call = cold-start proxy
    one_step()
    cold_start_s = time.perf_counter() - t_init0
    for _ in range(max(0, args.warmup - 1)):
        one_step()

    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if dist_on:
        import torch.distributed as dist

        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if dist_on:
        import torch.distributed as dist

        t = torch.tensor([elapsed], device=device if "cuda" in device else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        dist.barrier()

    n_gpus = world if dist_on else (1 if torch.cuda.is_available() else args.gpus or 1)
    total_images = args.steps * args.batch * (world if dist_on else 1)
    images_per_s = total_images / elapsed
    baseline_per_gpu = 1.0  # img/s/H100, reference best case

    if rank == 0:
        out = {
            "metric": "SDXL 1024px images/sec (whole node)",
            "value": round(images_per_s, 4),
            "unit": "images/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(images_per_s / (baseline_per_gpu * n_gpus), 4),
            "dtype": "bf16",
            "data": "synthetic prompts, random-init weights",
            "cold_start_s": round(cold_start_s, 2),
            "config": {
                "model": "sdxl-base-unet-2.6b" if not args.small else "sdxl-small-test",
                "global_batch": args.batch * (world if dist_on else 1),
                "image": f"{args.latent * 8}x{args.latent * 8}",
                "denoise_steps": args.denoise_steps,
                "parallelism": f"dp{n_gpus}",
                "hipgraph": not args.no_graph,
            },
        }
        print(json.dumps(out), flush=True)

    if dist_on:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(main())
