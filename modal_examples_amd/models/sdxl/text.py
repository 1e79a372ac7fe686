"""Synthetic prompt conditioning.

The benchmark contract (BASELINE.json) is synthetic prompts + random-init
weights — there is no network for the dual-CLIP text encoder checkpoints the
reference downloads (text_to_image.py:99-105).  This module produces
DETERMINISTIC pseudo-embeddings per prompt string (hash-seeded gaussians with
the real SDXL conditioning shapes), so caching/serving behavior matches the
real pipeline byte-for-byte in shape and dtype.
"""
from __future__ import annotations

import hashlib
from typing import List, Tuple

import torch


def encode_prompts(prompts: List[str], ctx_dim: int = 2048, pooled_dim: int = 1280,
                   seq_len: int = 77, device="cpu", dtype=torch.bfloat16
                   ) -> Tuple[torch.Tensor, torch.Tensor]:
    ctxs, pools = [], []
    for p in prompts:
        seed = int.from_bytes(hashlib.sha256(p.encode()).digest()[:8], "little") % (2**31)
        g = torch.Generator(device="cpu").manual_seed(seed)
        ctxs.append(torch.randn(seq_len, ctx_dim, generator=g))
        pools.append(torch.randn(pooled_dim, generator=g))
    ctx = torch.stack(ctxs).to(device=device, dtype=dtype)
    pooled = torch.stack(pools).to(device=device, dtype=dtype)
    return ctx, pooled


def fourier_time_ids(batch: int, fdim: int = 32, height: int = 1024,
                     width: int = 1024, device="cpu", dtype=torch.bfloat16):
    """SDXL micro-conditioning: (orig_h, orig_w, crop_t, crop_l, tgt_h, tgt_w)
    → per-value fourier features of dim ``fdim*2`` (256 for full SDXL)."""
    vals = torch.tensor([height, width, 0, 0, height, width], dtype=torch.float32)
    freqs = torch.exp(
        -torch.log(torch.tensor(10000.0)) * torch.arange(fdim, dtype=torch.float32) / fdim
    )
    emb = vals[:, None] * freqs[None]
    emb = torch.cat([emb.cos(), emb.sin()], dim=-1).flatten()  # [6 * 2fdim]
    return emb.to(device=device, dtype=dtype).expand(batch, -1).contiguous()
