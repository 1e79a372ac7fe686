#!/usr/bin/env python3
"""LoRA fine-tune step throughput (config 5): SDXL UNet rank-16 LoRA, bf16.

Reference anchor: A100-80GB Dreambooth run, 500 steps ≈ 10 min ≈ 0.83
steps/s at batch 3 / 512px (diffusers_lora_finetune.py:211,259-266).
Launchable under torchrun for DP-N (bucketed RCCL all-reduce overlap).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--batch", type=int, default=3)
    ap.add_argument("--resolution", type=int, default=512)
    ap.add_argument("--small", action="store_true")
    args = ap.parse_args()

    import torch

    from modal_examples_amd.models.sdxl.unet import UNetConfig
    from modal_examples_amd.train.dreambooth import LoRATrainer, TrainConfig

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = UNetConfig.small() if args.small or dev == "cpu" else UNetConfig.sdxl()
    t = LoRATrainer(
        cfg,
        TrainConfig(rank=16, batch_size=args.batch, resolution=args.resolution,
                    max_steps=10**9),
        device=dev, dtype=torch.bfloat16 if dev == "cuda" else torch.float32)
    for _ in range(args.warmup):
        t.train_step()
    if dev == "cuda":
        torch.cuda.synchronize()
    if t.world > 1:
        import torch.distributed as dist

        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = t.train_step()
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if t.world > 1:
        import torch.distributed as dist

        tt = torch.tensor([dt])
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        dt = float(tt)
    if t.rank_id == 0:
        sps = args.steps / dt
        print(json.dumps({
            "metric": "lora train steps/s",
            "value": round(sps, 3),
            "images_per_s": round(sps * args.batch * t.world, 2),
            "ms_per_step": round(dt / args.steps * 1e3, 1),
            "world": t.world,
            "batch_per_gpu": args.batch,
            "resolution": args.resolution,
            "last_loss": round(loss, 4),
            "vs_a100_baseline": round(sps / 0.83, 2),
        }))


if __name__ == "__main__":
    main()
