# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/dreambooth/lora_finetune.py", "--max-steps", "3"]
# ---
# # Dreambooth LoRA fine-tune (the canonical training example)
#
# Rank-16 LoRA over the SDXL UNet attention projections, bf16, with the fused
# AdamW kernel; gradients sync with bucketed RCCL all-reduce when launched
# multi-rank (torchrun).  Checkpoints land on a Volume (`volume.commit`), and
# inference loads them back after `volume.reload` — the reference's
# train→commit→serve flow.

import os

import modal_examples_amd as modal

app = modal.App("example-lora-finetune")

weights = modal.Volume.from_name("lora-weights", create_if_missing=True)


@app.function(gpu="mi355x:8", timeout=3600)
def train_distributed(max_steps: int = 3) -> str:
    """DP over every visible GPU: the reference's `accelerate launch`
    subprocess pattern (double process boundary) becomes torchrun — one rank
    per GPU, bucketed RCCL all-reduce inside the trainer."""
    import subprocess
    import sys

    import torch

    n = max(1, torch.cuda.device_count())
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
           "--master-port", "29561", "scripts/bench_train.py",
           "--steps", str(max_steps), "--warmup", "0"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=3000)
    print(r.stdout[-800:])
    assert r.returncode == 0, r.stderr[-800:]
    return f"trained on {n} rank(s)"


@app.function(gpu="mi355x", timeout=3600)
def train(max_steps: int = 3) -> str:
    import torch

    from modal_examples_amd.models.sdxl.unet import UNetConfig
    from modal_examples_amd.train.dreambooth import LoRATrainer, TrainConfig

    gpu = torch.cuda.is_available()
    trainer = LoRATrainer(
        UNetConfig.sdxl() if gpu else UNetConfig.small(),
        TrainConfig(rank=16 if gpu else 2, batch_size=3 if gpu else 1,
                    resolution=512 if gpu else 64, max_steps=max_steps,
                    checkpoint_every=max(1, max_steps // 2)),
        device="cuda" if gpu else "cpu",
        dtype=torch.bfloat16 if gpu else torch.float32,
        checkpoint_dir=str(weights.path),
    )
    trainer.load_checkpoint()  # resume if a previous run was interrupted
    trainer.train(max_steps=max_steps)
    weights.commit()
    return os.path.join(str(weights.path), "last.ckpt")


@app.function(gpu="mi355x")
def sample_with_lora(ckpt_path: str) -> int:
    import torch

    from modal_examples_amd.models.sdxl.unet import UNetConfig, UNetXL
    from modal_examples_amd.train.lora import apply_lora, load_lora_state

    weights.reload()
    gpu = torch.cuda.is_available()
    cfg = UNetConfig.sdxl() if gpu else UNetConfig.small()
    net = UNetXL(cfg).to("cuda" if gpu else "cpu",
                        torch.bfloat16 if gpu else torch.float32)
    apply_lora(net, rank=16 if gpu else 2)
    ck = torch.load(ckpt_path, map_location="cpu", weights_only=False)
    load_lora_state(net, ck["lora"])
    n = sum(v.numel() for v in ck["lora"].values())
    print(f"loaded LoRA checkpoint from step {ck['step']} ({n} params)")
    return ck["step"]


@app.local_entrypoint()
def main(max_steps: int = 3):
    ckpt = train.remote(max_steps)
    step = sample_with_lora.remote(ckpt)
    print(f"trained to step {step}, weights at {ckpt}")
