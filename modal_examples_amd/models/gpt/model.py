"""nanoGPT-class causal LM for the hyperparameter-sweep example.

Capability mirror of the reference's sweep target
(06_gpu_and_ml/hyperparameter-sweep/hp_sweep_gpt.py + src/model.py: a small
GPT trained from scratch, checkpointed to a Volume, early-stopped, resumed).
Hot ops on gfx950 kernels: causal flash attention (head_dim 64), LayerNorm,
fused sampling for generation; GEMMs via hipBLASLt.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from ...ops import functional as OF
from ..sdxl.layers import LayerNormK


@dataclass
class GPTConfig:
    vocab_size: int = 256      # byte-level (tokenizer-free, like char nanoGPT)
    block_size: int = 256
    n_layer: int = 6
    n_head: int = 6
    n_embd: int = 384
    dropout: float = 0.0

    @staticmethod
    def gpt2_small() -> "GPTConfig":
        return GPTConfig(vocab_size=50304, block_size=1024, n_layer=12,
                         n_head=12, n_embd=768)


class Block(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.ln1 = LayerNormK(cfg.n_embd)
        self.qkv = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.proj = nn.Linear(cfg.n_embd, cfg.n_embd)
        self.ln2 = LayerNormK(cfg.n_embd)
        self.fc = nn.Linear(cfg.n_embd, 4 * cfg.n_embd)
        self.fc_proj = nn.Linear(4 * cfg.n_embd, cfg.n_embd)
        self.h = cfg.n_head
        self.d = cfg.n_embd // cfg.n_head

    def forward(self, x):
        B, S, C = x.shape
        qkv = self.qkv(self.ln1(x)).view(B, S, 3, self.h, self.d)
        o = OF.attention_qkv(qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2], causal=True)
        x = x + self.proj(o)
        h = torch.nn.functional.gelu(self.fc(self.ln2(x)), approximate="tanh")
        return x + self.fc_proj(h)


class GPT(nn.Module):
    def __init__(self, cfg: GPTConfig = None):
        super().__init__()
        cfg = cfg or GPTConfig()
        self.cfg = cfg
        self.tok = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.pos = nn.Embedding(cfg.block_size, cfg.n_embd)
        self.blocks = nn.ModuleList([Block(cfg) for _ in range(cfg.n_layer)])
        self.ln_f = LayerNormK(cfg.n_embd)
        self.head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.head.weight = self.tok.weight  # weight tying
        # GPT-2-style init: torch's default N(0,1) embeddings give ±15-unit
        # logits at init, which makes sampling deterministic (argmax) — bad
        # for anything exploring from the prior (e.g. RL rollouts).
        self.apply(self._init_weights)

    @staticmethod
    def _init_weights(m: nn.Module):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, mean=0.0, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, idx: torch.Tensor, targets: torch.Tensor = None):
        B, S = idx.shape
        pos = torch.arange(S, device=idx.device)
        x = self.tok(idx) + self.pos(pos)
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x)
        logits = self.head(x)
        if targets is None:
            return logits, None
        loss = torch.nn.functional.cross_entropy(
            logits.view(-1, logits.size(-1)).float(), targets.reshape(-1))
        return logits, loss

    @torch.no_grad()
    def generate(self, idx: torch.Tensor, max_new_tokens: int,
                 temperature: float = 1.0, seed: int = 0):
        for i in range(max_new_tokens):
            ctx = idx[:, -self.cfg.block_size:]
            logits, _ = self(ctx)
            nxt = OF.sample(logits[:, -1].float(), temperature, seed=seed + i)
            idx = torch.cat([idx, nxt.long().view(-1, 1)], dim=1)
        return idx
