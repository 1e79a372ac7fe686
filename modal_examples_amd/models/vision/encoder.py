"""ViT-class vision encoder on the gfx950 attention kernel (K12's engine role).

Reference roles covered: the vision half of the VLM / pdf-RAG examples
(06_gpu_and_ml/llm-serving/chat_with_pdf_vision.py — ColPali-style page-image
embeddings scored by late interaction; sglang_vlm.py — image encoder feeding
an LLM).  MI355X mapping: patch-embed is one GEMM (K2), every block's
attention runs the K1 flash kernel via the transpose-free BSHD path, MLPs are
hipBLASLt GEMMs.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from ...ops import functional as OF


@dataclass
class ViTConfig:
    image_size: int = 448
    patch: int = 16
    hidden: int = 1024
    heads: int = 16      # head_dim 64 — K1's D=64 MFMA path
    depth: int = 24
    embed_dim: int = 128  # multi-vector projection width (ColPali-style)

    @staticmethod
    def base() -> "ViTConfig":
        return ViTConfig()

    @staticmethod
    def small_test() -> "ViTConfig":
        return ViTConfig(image_size=64, patch=16, hidden=128, heads=2, depth=2,
                         embed_dim=32)


class ViTBlock(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        h = cfg.hidden
        self.heads = cfg.heads
        self.hd = h // cfg.heads
        self.norm1 = nn.LayerNorm(h)
        self.qkv = nn.Linear(h, 3 * h)  # fused QKV GEMM (K2)
        self.proj = nn.Linear(h, h)
        self.norm2 = nn.LayerNorm(h)
        self.mlp = nn.Sequential(nn.Linear(h, 4 * h), nn.GELU(approximate="tanh"),
                                 nn.Linear(4 * h, h))

    def forward(self, x):
        B, S, h = x.shape
        # LayerNorm runs in the module dtype: affine weights are bf16 after
        # .to(dtype), and GPU layer_norm rejects float-input/bf16-weight
        qkv = self.qkv(self.norm1(x)).view(B, S, 3, self.heads, self.hd)
        o = OF.attention_qkv(qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2])
        x = x + self.proj(o)
        return x + self.mlp(self.norm2(x))


class VisionEncoder(nn.Module):
    """Image [B,3,H,W] → per-patch embeddings [B, n_patches, embed_dim],
    L2-normalized (late-interaction / MaxSim ready)."""

    def __init__(self, cfg: ViTConfig = None):
        super().__init__()
        cfg = cfg or ViTConfig.base()
        self.cfg = cfg
        n = (cfg.image_size // cfg.patch) ** 2
        self.patch_embed = nn.Conv2d(3, cfg.hidden, cfg.patch, stride=cfg.patch)
        self.pos = nn.Parameter(torch.zeros(1, n, cfg.hidden))
        nn.init.normal_(self.pos, std=0.02)
        self.blocks = nn.ModuleList(ViTBlock(cfg) for _ in range(cfg.depth))
        self.norm = nn.LayerNorm(cfg.hidden)
        self.out_proj = nn.Linear(cfg.hidden, cfg.embed_dim)

    @torch.no_grad()
    def embed(self, images: torch.Tensor) -> torch.Tensor:
        x = self.patch_embed(images).flatten(2).transpose(1, 2)
        x = x + self.pos[:, : x.shape[1]]
        for blk in self.blocks:
            x = blk(x)
        e = self.out_proj(self.norm(x))
        return torch.nn.functional.normalize(e.float(), dim=-1)


def maxsim(query: torch.Tensor, pages: torch.Tensor) -> torch.Tensor:
    """ColPali late-interaction score: for each query vector take the best
    page-patch match, sum over query vectors.  query [Q,D], pages [P,N,D]
    (L2-normalized) → scores [P]."""
    sim = torch.einsum("qd,pnd->pqn", query, pages)
    return sim.max(dim=-1).values.sum(dim=-1)
