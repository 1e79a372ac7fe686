"""Flux-class MMDiT: the second diffusion architecture (reference trigger:
06_gpu_and_ml/stable_diffusion/flux.py:111-273 — FLUX.1-schnell, a rectified
flow-matching MMDiT).

Built MI355X-first, not a diffusers port: joint image+text attention runs on
the gfx950 flash kernel (K1, D=128 path), QKV projections are single fused
GEMMs (K2), modulation/gating uses the fused silu kernel, and the denoise
loop is hipGraph-captured by the pipeline (K10's role — no torch.compile).

Shape follows the schnell class: double-stream blocks (separate img/txt
streams, joint attention) then single-stream blocks, adaLN modulation from
(timestep, pooled-text) embeddings, 2x2-packed latent patches.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from ...ops import functional as OF
from ..sdxl.layers import timestep_embedding


@dataclass
class MMDiTConfig:
    hidden: int = 3072
    heads: int = 24           # head_dim 128 — the K1/K7 D=128 MFMA path
    double_blocks: int = 19
    single_blocks: int = 38
    ctx_dim: int = 4096       # T5-class conditioning width
    pooled_dim: int = 768
    txt_len: int = 512
    latent_channels: int = 4  # packed 2x2 -> 16 per token
    mlp_ratio: float = 4.0

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads

    @staticmethod
    def schnell() -> "MMDiTConfig":
        return MMDiTConfig()

    @staticmethod
    def small() -> "MMDiTConfig":
        """Test-sized variant (same topology)."""
        return MMDiTConfig(hidden=256, heads=2, double_blocks=2,
                           single_blocks=4, ctx_dim=128, pooled_dim=64,
                           txt_len=16)


class Modulation(nn.Module):
    """adaLN: vec -> per-block (shift, scale, gate) sets."""

    def __init__(self, hidden: int, n_sets: int):
        super().__init__()
        self.lin = nn.Linear(hidden, 3 * n_sets * hidden)
        self.n_sets = n_sets

    def forward(self, vec):
        out = self.lin(nn.functional.silu(vec))
        return out.view(vec.shape[0], self.n_sets, 3, -1).unbind(2)  # shift/scale/gate


def _mod(x, shift, scale, i):
    return x * (1 + scale[:, i, None]) + shift[:, i, None]


class DoubleBlock(nn.Module):
    """Separate img/txt streams; ONE joint attention over the concat seq."""

    def __init__(self, cfg: MMDiTConfig):
        super().__init__()
        h = cfg.hidden
        self.heads, self.hd = cfg.heads, cfg.head_dim
        self.img_mod = Modulation(h, 2)
        self.txt_mod = Modulation(h, 2)
        self.img_norm1 = nn.LayerNorm(h, elementwise_affine=False)
        self.txt_norm1 = nn.LayerNorm(h, elementwise_affine=False)
        self.img_qkv = nn.Linear(h, 3 * h)
        self.txt_qkv = nn.Linear(h, 3 * h)
        self.img_proj = nn.Linear(h, h)
        self.txt_proj = nn.Linear(h, h)
        self.img_norm2 = nn.LayerNorm(h, elementwise_affine=False)
        self.txt_norm2 = nn.LayerNorm(h, elementwise_affine=False)
        inner = int(h * cfg.mlp_ratio)
        self.img_mlp = nn.Sequential(nn.Linear(h, inner), nn.GELU(approximate="tanh"),
                                     nn.Linear(inner, h))
        self.txt_mlp = nn.Sequential(nn.Linear(h, inner), nn.GELU(approximate="tanh"),
                                     nn.Linear(inner, h))

    def forward(self, img, txt, vec):
        B, Si, _ = img.shape
        St = txt.shape[1]
        ish, isc, igt = self.img_mod(vec)
        tsh, tsc, tgt = self.txt_mod(vec)
        iq = _mod(self.img_norm1(img.float()).to(img.dtype), ish, isc, 0)
        tq = _mod(self.txt_norm1(txt.float()).to(txt.dtype), tsh, tsc, 0)
        iqkv = self.img_qkv(iq).view(B, Si, 3, self.heads, self.hd)
        tqkv = self.txt_qkv(tq).view(B, St, 3, self.heads, self.hd)
        q = torch.cat([tqkv[:, :, 0], iqkv[:, :, 0]], dim=1)
        k = torch.cat([tqkv[:, :, 1], iqkv[:, :, 1]], dim=1)
        v = torch.cat([tqkv[:, :, 2], iqkv[:, :, 2]], dim=1)
        o = OF.attention_qkv(q, k, v)  # [B, St+Si, h]
        txt = txt + tgt[:, 0, None] * self.txt_proj(o[:, :St])
        img = img + igt[:, 0, None] * self.img_proj(o[:, St:])
        im = _mod(self.img_norm2(img.float()).to(img.dtype), ish, isc, 1)
        tm = _mod(self.txt_norm2(txt.float()).to(txt.dtype), tsh, tsc, 1)
        img = img + igt[:, 1, None] * self.img_mlp(im)
        txt = txt + tgt[:, 1, None] * self.txt_mlp(tm)
        return img, txt


class SingleBlock(nn.Module):
    """Fused stream: one norm, parallel attention + MLP, one output proj
    (the flux single-block shape — fewer GEMM launches than sequential)."""

    def __init__(self, cfg: MMDiTConfig):
        super().__init__()
        h = cfg.hidden
        self.heads, self.hd = cfg.heads, cfg.head_dim
        inner = int(h * cfg.mlp_ratio)
        self.mod = Modulation(h, 1)
        self.norm = nn.LayerNorm(h, elementwise_affine=False)
        self.qkv_mlp = nn.Linear(h, 3 * h + inner)  # fused QKV+MLP-in GEMM
        self.out = nn.Linear(h + inner, h)          # fused attn+MLP-out GEMM
        self.inner = inner

    def forward(self, x, vec):
        B, S, h = x.shape
        sh, sc, gt = self.mod(vec)
        xm = _mod(self.norm(x.float()).to(x.dtype), sh, sc, 0)
        qkv_m = self.qkv_mlp(xm)
        qkv = qkv_m[..., : 3 * h].view(B, S, 3, self.heads, self.hd)
        o = OF.attention_qkv(qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2])
        mlp = nn.functional.gelu(qkv_m[..., 3 * h:], approximate="tanh")
        return x + gt[:, 0, None] * self.out(torch.cat([o, mlp], dim=-1))


class MMDiT(nn.Module):
    def __init__(self, cfg: MMDiTConfig = None):
        super().__init__()
        cfg = cfg or MMDiTConfig.schnell()
        self.cfg = cfg
        h = cfg.hidden
        self.img_in = nn.Linear(cfg.latent_channels * 4, h)  # 2x2 patch pack
        self.txt_in = nn.Linear(cfg.ctx_dim, h)
        self.time_in = nn.Sequential(nn.Linear(256, h), nn.SiLU(), nn.Linear(h, h))
        self.vec_in = nn.Sequential(nn.Linear(cfg.pooled_dim, h), nn.SiLU(),
                                    nn.Linear(h, h))
        self.pos_img = nn.Parameter(torch.zeros(1, 64 * 64, h))  # up to 128x128 latent
        self.pos_txt = nn.Parameter(torch.zeros(1, cfg.txt_len, h))
        nn.init.normal_(self.pos_img, std=0.02)
        nn.init.normal_(self.pos_txt, std=0.02)
        self.dblocks = nn.ModuleList(DoubleBlock(cfg) for _ in range(cfg.double_blocks))
        self.sblocks = nn.ModuleList(SingleBlock(cfg) for _ in range(cfg.single_blocks))
        self.final_mod = nn.Linear(h, 2 * h)
        self.final_norm = nn.LayerNorm(h, elementwise_affine=False)
        self.final_out = nn.Linear(h, cfg.latent_channels * 4)

    @staticmethod
    def pack(x):
        """[B,C,H,W] -> [B, H/2*W/2, C*4] (2x2 patches as tokens)."""
        B, C, H, W = x.shape
        x = x.view(B, C, H // 2, 2, W // 2, 2)
        return x.permute(0, 2, 4, 1, 3, 5).reshape(B, (H // 2) * (W // 2), C * 4)

    @staticmethod
    def unpack(t, C, H, W):
        B = t.shape[0]
        x = t.view(B, H // 2, W // 2, C, 2, 2)
        return x.permute(0, 3, 1, 4, 2, 5).reshape(B, C, H, W)

    def forward(self, x, t, ctx, pooled):
        """x [B,C,H,W] latent, t [B] (flow time in [0,1]*1000), ctx
        [B,txt_len,ctx_dim], pooled [B,pooled_dim] -> velocity [B,C,H,W]."""
        B, C, H, W = x.shape
        dt = x.dtype
        img = self.img_in(self.pack(x))
        img = img + self.pos_img[:, : img.shape[1]]
        txt = self.txt_in(ctx) + self.pos_txt[:, : ctx.shape[1]]
        vec = self.time_in(timestep_embedding(t, 256).to(dt)) + self.vec_in(pooled)
        for blk in self.dblocks:
            img, txt = blk(img, txt, vec)
        s = torch.cat([txt, img], dim=1)
        for blk in self.sblocks:
            s = blk(s, vec)
        img = s[:, txt.shape[1]:]
        m = self.final_mod(nn.functional.silu(vec))
        shift, scale = m.chunk(2, dim=-1)
        img = self.final_norm(img.float()).to(dt) * (1 + scale[:, None]) + shift[:, None]
        return self.unpack(self.final_out(img), C, H, W)
