"""LoRA adapters + fused-AdamW optimizer (K9).

Reference behavior spec: Dreambooth LoRA fine-tune of the diffusion
transformer's attention projections, rank 16, bf16 compute
(diffusers_lora_finetune.py:264-339).  Adapters keep f32 master weights so the
fused AdamW kernel updates exactly; the base model stays frozen bf16.
"""
from __future__ import annotations

from typing import Dict, Iterable, List

import torch
import torch.nn as nn

from ..ops import functional as OF

DEFAULT_TARGETS = ("qkv", "q", "kv", "out", "proj_in", "proj_out")


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Linear, rank: int = 16, alpha: float = 16.0):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        self.rank = rank
        self.scale = alpha / rank
        dev = base.weight.device
        self.lora_a = nn.Parameter(
            torch.randn(rank, base.in_features, dtype=torch.float32,
                        device=dev) * 0.01)
        self.lora_b = nn.Parameter(
            torch.zeros(base.out_features, rank, dtype=torch.float32,
                        device=dev))

    def forward(self, x):
        y = self.base(x)
        lx = (x.float() @ self.lora_a.T @ self.lora_b.T) * self.scale
        return y + lx.to(y.dtype)


def apply_lora(module: nn.Module, rank: int = 16, alpha: float = 16.0,
               targets: Iterable[str] = DEFAULT_TARGETS) -> List[str]:
    """Replace attention-projection Linears with LoRA wrappers; returns the
    qualified names wrapped."""
    wrapped = []
    targets = set(targets)
    for name, child in list(module.named_children()):
        if isinstance(child, nn.Linear) and name in targets:
            setattr(module, name, LoRALinear(child, rank, alpha))
            wrapped.append(name)
        else:
            wrapped.extend(f"{name}.{w}" for w in apply_lora(child, rank, alpha, targets))
    return wrapped


def lora_parameters(module: nn.Module) -> List[nn.Parameter]:
    return [p for n, p in module.named_parameters()
            if p.requires_grad and ("lora_a" in n or "lora_b" in n)]


def lora_state_dict(module: nn.Module) -> Dict[str, torch.Tensor]:
    return {n: p.detach().cpu() for n, p in module.named_parameters()
            if "lora_a" in n or "lora_b" in n}


def load_lora_state(module: nn.Module, state: Dict[str, torch.Tensor]):
    own = {n: p for n, p in module.named_parameters()
           if "lora_a" in n or "lora_b" in n}
    for n, t in state.items():
        own[n].data.copy_(t.to(own[n].device))


class FusedAdamW:
    """Optimizer driving the gfx950 fused AdamW kernel (K9); torch-equivalent
    semantics (bias correction per step, decoupled weight decay)."""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.01):
        self.params = [p for p in params if p.requires_grad]
        self.lr, self.betas, self.eps, self.wd = lr, betas, eps, weight_decay
        self.step_count = 0
        self.m = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]
        self.v = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        for p, m, v in zip(self.params, self.m, self.v):
            if p.grad is None:
                continue
            OF.adamw_step(p.data, p.grad.to(p.dtype), m, v, self.lr,
                          self.betas[0], self.betas[1], self.eps, self.wd,
                          self.step_count)

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def state_dict(self):
        return {"step": self.step_count,
                "m": [t.cpu() for t in self.m],
                "v": [t.cpu() for t in self.v]}

    def load_state_dict(self, st):
        self.step_count = st["step"]
        for dst, src in zip(self.m, st["m"]):
            dst.copy_(src.to(dst.device))
        for dst, src in zip(self.v, st["v"]):
            dst.copy_(src.to(dst.device))
