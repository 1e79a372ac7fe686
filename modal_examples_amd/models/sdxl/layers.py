"""Shared MI355X-native building blocks for the diffusion models.

Every hot op routes to the gfx950 kernels in ops/functional.py; plain GEMMs and
convolutions go through torch (hipBLASLt / MIOpen per the library-GEMM rule).
QKV projections are single fused GEMMs (K2 fusion, reference trigger:
stable_diffusion/flux.py:236-237 `fuse_qkv_projections`).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ...ops import functional as OF


class _F32ParamCache:
    """Norm params live in whatever dtype the module was cast to; the kernels
    want f32.  In eval mode the f32 copies are cached (profiling showed
    thousands of tiny bf16→f32 casts per SDXL step); training recomputes so
    autograd sees the live parameters."""

    def _f32(self, name: str):
        p = getattr(self, name)
        if self.training or p.requires_grad and torch.is_grad_enabled():
            return p.float()
        cache = self.__dict__.setdefault("_f32_cache", {})
        ent = cache.get(name)
        if ent is None or ent[1] is not p or ent[2] != p._version:
            ent = (p.detach().float(), p, p._version)
            cache[name] = ent
        return ent[0]


class LayerNormK(nn.Module, _F32ParamCache):
    """LayerNorm over last dim via the gfx950 kernel (f32 affine params)."""

    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):
        return OF.layernorm(x, self._f32("weight"), self._f32("bias"), self.eps)


class RMSNormK(nn.Module, _F32ParamCache):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return OF.rmsnorm(x, self._f32("weight"), self.eps)


class GroupNormSiLU(nn.Module, _F32ParamCache):
    """Fused GroupNorm(+SiLU) NCHW via the gfx950 kernel."""

    def __init__(self, channels: int, groups: int = 32, eps: float = 1e-5,
                 silu: bool = True):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))
        self.groups = groups
        self.eps = eps
        self.silu = silu

    def forward(self, x):
        return OF.groupnorm_silu(x, self._f32("weight"), self._f32("bias"),
                                 self.groups, self.eps, self.silu)


class Conv3x3(nn.Conv2d):
    """3x3 stride-1 pad-1 conv on the hand NCHW implicit-GEMM kernel (K3).

    Drop-in for nn.Conv2d(c_in, c_out, 3, padding=1): same parameters and
    state_dict.  Inference on GPU bf16 runs the MFMA kernel (with an optional
    fused residual add); training / CPU / odd dtypes use F.conv2d so autograd
    and hermetic tests stay correct.  The [9,Kpad,C16] weight repack is cached
    against the parameter version."""

    def __init__(self, c_in: int, c_out: int):
        super().__init__(c_in, c_out, 3, padding=1)

    def _packed(self):
        w = self.weight
        ent = self.__dict__.get("_wr_cache")
        if ent is None or ent[1] is not w or ent[2] != w._version:
            wr = OF.repack_conv3x3_weight(w.detach())
            b = (self.bias.detach().float().contiguous() if self.bias is not None
                 else torch.zeros(w.shape[0], device=w.device))
            ent = ((wr, b), w, w._version)
            self.__dict__["_wr_cache"] = ent
        return ent[0]

    def _use_kernel(self, x) -> bool:
        if not x.is_cuda or x.dtype is not torch.bfloat16 or self.training:
            return False
        if torch.is_grad_enabled() and (x.requires_grad or
                                        self.weight.requires_grad):
            return False
        return True

    def forward_gn(self, x, gn, residual=None, upsample: bool = False):
        """Fused GroupNorm+SiLU -> conv (gn: a GroupNormSiLU module with
        silu=True).  Falls back to gn(x) then conv off the kernel path."""
        if self._use_kernel(x) and gn.silu:
            wr, b = self._packed()
            return OF.conv3x3_gn(x, wr, b, self.out_channels,
                                 gn._f32("weight"), gn._f32("bias"),
                                 gn.groups, gn.eps, residual=residual,
                                 raw_weight=self.weight, upsample=upsample)
        return self.forward(gn(x), residual=residual, upsample=upsample)

    def forward(self, x, residual=None, upsample: bool = False):
        if self._use_kernel(x):
            wr, b = self._packed()
            return OF.conv3x3(x, wr, b, self.out_channels, residual=residual,
                              raw_weight=self.weight, upsample=upsample)
        if upsample:
            x = nn.functional.interpolate(x, scale_factor=2.0, mode="nearest")
        y = nn.functional.conv2d(x, self.weight, self.bias, padding=1)
        return y if residual is None else y + residual


class Conv1x1(nn.Conv2d):
    """1x1 conv (skip projections).  MIOpen's 1x1 path is a direct GEMM with
    no layout transposes, so the library call is already the fast path — a
    broadcast `w @ x` matmul was measured to MATERIALIZE the expanded weight
    (≈20 ms of copies per VAE decode) and was reverted."""

    def __init__(self, c_in: int, c_out: int):
        super().__init__(c_in, c_out, 1)


class SelfAttention(nn.Module):
    """Fused-QKV self-attention over [B, S, C] → gfx950 flash kernel."""

    def __init__(self, dim: int, head_dim: int = 64):
        super().__init__()
        assert dim % head_dim == 0
        self.heads = dim // head_dim
        self.head_dim = head_dim
        self.qkv = nn.Linear(dim, 3 * dim, bias=False)
        self.out = nn.Linear(dim, dim, bias=True)

    def forward(self, x):
        B, S, C = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.heads, self.head_dim)
        # transpose-free: the kernel reads the fused-projection slices directly
        o = OF.attention_qkv(qkv[:, :, 0], qkv[:, :, 1], qkv[:, :, 2])
        return self.out(o)


class CrossAttention(nn.Module):
    """Cross-attention: q from x, fused kv from context (text conditioning)."""

    def __init__(self, dim: int, ctx_dim: int, head_dim: int = 64):
        super().__init__()
        self.heads = dim // head_dim
        self.head_dim = head_dim
        self.q = nn.Linear(dim, dim, bias=False)
        self.kv = nn.Linear(ctx_dim, 2 * dim, bias=False)
        self.out = nn.Linear(dim, dim, bias=True)

    def forward(self, x, ctx):
        B, S, C = x.shape
        Sk = ctx.shape[1]
        q = self.q(x).view(B, S, self.heads, self.head_dim)
        kv = self.kv(ctx).view(B, Sk, 2, self.heads, self.head_dim)
        o = OF.attention_qkv(q, kv[:, :, 0], kv[:, :, 1])
        return self.out(o)


class GEGLUFeedForward(nn.Module):
    """x → Linear(2*4c) → gelu(a)*b (fused kernel) → Linear(c)."""

    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        inner = dim * mult
        self.proj_in = nn.Linear(dim, 2 * inner, bias=True)
        self.proj_out = nn.Linear(inner, dim, bias=True)
        self.inner = inner

    def forward(self, x):
        # fused kernel reads both GEGLU halves of the projection in place
        return self.proj_out(OF.glu_fused(self.proj_in(x), gelu=True))


class TransformerBlock(nn.Module):
    """ln→self-attn→ln→cross-attn→ln→GEGLU-FF, pre-norm residuals (the
    diffusers BasicTransformerBlock shape exercised at text_to_image.py:114)."""

    def __init__(self, dim: int, ctx_dim: int, head_dim: int = 64):
        super().__init__()
        self.norm1 = LayerNormK(dim)
        self.attn1 = SelfAttention(dim, head_dim)
        self.norm2 = LayerNormK(dim)
        self.attn2 = CrossAttention(dim, ctx_dim, head_dim)
        self.norm3 = LayerNormK(dim)
        self.ff = GEGLUFeedForward(dim)

    def forward(self, x, ctx):
        x = x + self.attn1(self.norm1(x))
        x = x + self.attn2(self.norm2(x), ctx)
        x = x + self.ff(self.norm3(x))
        return x


def timestep_embedding(t: torch.Tensor, dim: int, max_period: float = 10000.0):
    """Sinusoidal timestep embedding, f32 (host-side trig, tiny)."""
    half = dim // 2
    freqs = torch.exp(
        -math.log(max_period) * torch.arange(half, dtype=torch.float32, device=t.device) / half
    )
    args = t.float()[:, None] * freqs[None]
    return torch.cat([torch.cos(args), torch.sin(args)], dim=-1)
