"""Dreambooth-style LoRA fine-tune of the SDXL UNet, DP over RCCL/xGMI.

The MI355X-native equivalent of the reference's canonical training config
(diffusers_lora_finetune.py:264-343: accelerate-launched LoRA training, bf16,
rank 16, then volume.commit of the weights).  Here: rank-per-GPU data
parallelism with bucketed gradient all-reduce overlapped with backward
(parallel/ddp.py), the fused AdamW kernel for the update, and checkpoint
save/resume with the step counter (the long-training.py:186-214 resume
semantics).
"""
from __future__ import annotations

import math
import os
import time
from dataclasses import dataclass
from typing import Optional

import torch

from ..models.sdxl.pipeline import euler_sigmas
from ..models.sdxl.text import encode_prompts, fourier_time_ids
from ..models.sdxl.unet import UNetConfig, UNetXL
from ..parallel.collectives import broadcast_module, init_distributed
from ..parallel.ddp import GradReducer
from .lora import FusedAdamW, apply_lora, load_lora_state, lora_parameters, lora_state_dict


@dataclass
class TrainConfig:
    rank: int = 16
    alpha: float = 16.0
    lr: float = 1e-4
    batch_size: int = 3          # per GPU (reference: batch 3 @ 512px)
    resolution: int = 512        # latent 64
    max_steps: int = 500
    checkpoint_every: int = 100
    seed: int = 0


class LoRATrainer:
    def __init__(self, unet_cfg: Optional[UNetConfig] = None,
                 train_cfg: Optional[TrainConfig] = None,
                 device: str = "cuda", dtype=torch.bfloat16,
                 checkpoint_dir: Optional[str] = None):
        self.tc = train_cfg or TrainConfig()
        self.rank_id, self.world, self.local = init_distributed()
        if device == "cuda" and torch.cuda.is_available():
            device = f"cuda:{self.local}"
        self.device = torch.device(device)
        self.dtype = dtype
        torch.manual_seed(self.tc.seed)  # same base weights on every rank
        cfg = unet_cfg or UNetConfig.sdxl()
        self.cfg = cfg
        with torch.device(self.device):
            self.unet = UNetXL(cfg).to(self.device, dtype)
        apply_lora(self.unet, self.tc.rank, self.tc.alpha)
        self.unet.train()
        broadcast_module(self.unet, src=0)
        self.params = lora_parameters(self.unet)
        self.reducer = GradReducer(self.params)
        self.opt = FusedAdamW(self.params, lr=self.tc.lr)
        self.step_count = 0
        self.ckpt_dir = checkpoint_dir
        sig, _ = euler_sigmas(1000)
        self.all_sigmas = sig[:-1].to(self.device)
        self.latent = self.tc.resolution // 8
        self._gen = torch.Generator(device="cpu").manual_seed(
            self.tc.seed * 1000 + self.rank_id + 1)  # DIFFERENT data per rank

    # ---------------- data (synthetic: random latents + hashed prompts) ----

    def _batch(self):
        b = self.tc.batch_size
        x0 = torch.randn(b, 4, self.latent, self.latent, generator=self._gen
                         ).to(self.device, self.dtype)
        ctx, pooled = encode_prompts(
            [f"subject-{int(torch.randint(0, 8, (1,), generator=self._gen))}"
             for _ in range(b)],
            self.cfg.ctx_dim, self.cfg.pooled_dim, device=self.device,
            dtype=self.dtype)
        tid = fourier_time_ids(b, self.cfg.fourier_dim, self.tc.resolution,
                               self.tc.resolution, device=self.device,
                               dtype=self.dtype)
        add = torch.cat([pooled, tid], dim=-1)
        return x0, ctx, add

    # ---------------- one training step ----------------

    def train_step(self) -> float:
        x0, ctx, add = self._batch()
        b = x0.shape[0]
        t_idx = torch.randint(0, 1000, (b,), generator=self._gen).to(self.device)
        sigma = self.all_sigmas[t_idx].view(b, 1, 1, 1).to(self.dtype)
        eps = torch.randn(x0.shape, generator=self._gen).to(self.device, self.dtype)
        x_t = x0 + sigma * eps
        c_in = (1.0 / (sigma.float() ** 2 + 1.0).sqrt()).to(self.dtype)
        pred = self.unet(x_t * c_in, t_idx.float(), ctx, add)
        loss = torch.nn.functional.mse_loss(pred.float(), eps.float())
        loss.backward()
        self.reducer.finish()
        self.opt.step()
        self.opt.zero_grad()
        self.step_count += 1
        return float(loss.detach())

    # ---------------- checkpointing (long-training resume semantics) -------

    def save_checkpoint(self, path: Optional[str] = None):
        if self.rank_id != 0:
            return
        path = path or os.path.join(self.ckpt_dir, "last.ckpt")
        os.makedirs(os.path.dirname(path), exist_ok=True)
        torch.save({
            "step": self.step_count,
            "lora": lora_state_dict(self.unet),
            "opt": self.opt.state_dict(),
        }, path)

    def load_checkpoint(self, path: Optional[str] = None) -> bool:
        path = path or (os.path.join(self.ckpt_dir, "last.ckpt") if self.ckpt_dir else None)
        if not path or not os.path.exists(path):
            return False
        ck = torch.load(path, map_location=self.device, weights_only=False)
        load_lora_state(self.unet, ck["lora"])
        self.opt.load_state_dict(ck["opt"])
        self.step_count = ck["step"]
        return True

    def train(self, max_steps: Optional[int] = None, log_every: int = 25):
        max_steps = max_steps or self.tc.max_steps
        t0 = time.monotonic()
        while self.step_count < max_steps:
            loss = self.train_step()
            if self.step_count % log_every == 0 and self.rank_id == 0:
                dt = time.monotonic() - t0
                ips = self.step_count * self.tc.batch_size * self.world / max(dt, 1e-9)
                print(f"[rank0] step {self.step_count} loss {loss:.4f} "
                      f"({ips:.1f} img/s whole-job)", flush=True)
            if self.ckpt_dir and self.step_count % self.tc.checkpoint_every == 0:
                self.save_checkpoint()
        if self.ckpt_dir:
            self.save_checkpoint()
