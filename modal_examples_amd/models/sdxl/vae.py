"""SDXL-class VAE decoder (K3): latent [B,4,128,128] → RGB [B,3,1024,1024].

Architecture mirrors the SDXL AutoencoderKL decoder the reference's pipelines
call (trigger: text_to_image.py:114 `pipe(...)` → vae.decode; flux.py:259-261
compiles exactly this module).  Channels (512,512,256,128), 3 resnets per
level, nearest-2x upsample, mid-block single-head attention over 128² tokens.

MI355X mapping: GroupNorm+SiLU → fused gfx950 kernel; convs → MIOpen;
mid attention (1 head × 512 dim) → chunked hipBLASLt matmul+softmax (head_dim
512 is outside the MFMA flash kernel's D∈{64,128}; one layer, ~0.5 GFLOP-ms).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .layers import Conv1x1, Conv3x3, GroupNormSiLU


class VAEResnet(nn.Module):
    def __init__(self, c_in, c_out):
        super().__init__()
        self.norm1 = GroupNormSiLU(c_in)
        self.conv1 = Conv3x3(c_in, c_out)
        self.norm2 = GroupNormSiLU(c_out)
        self.conv2 = Conv3x3(c_out, c_out)
        self.skip = Conv1x1(c_in, c_out) if c_in != c_out else nn.Identity()

    def forward(self, x):
        # conv2 fuses the skip add (K3 epilogue).  NOTE measured negative
        # result: fusing GN+SiLU INTO conv staging (conv3x3_gn) regressed the
        # step 13.3->9.4 img/s — every output-k-tile re-reads the input, so
        # the fused exp recomputes K/64 times per element.  The separate
        # two-pass GN kernel stays the fast path.
        h = self.conv1(self.norm1(x))
        return self.conv2(self.norm2(h), residual=self.skip(x))


class VAEMidAttention(nn.Module):
    """Single-head 512-dim attention, chunked over queries."""

    def __init__(self, c, chunk=2048):
        super().__init__()
        self.norm = GroupNormSiLU(c, silu=False)
        self.q = nn.Linear(c, c)
        self.k = nn.Linear(c, c)
        self.v = nn.Linear(c, c)
        self.out = nn.Linear(c, c)
        self.chunk = chunk
        self.scale = c ** -0.5

    def forward(self, x):
        B, C, H, W = x.shape
        h = self.norm(x).permute(0, 2, 3, 1).reshape(B, H * W, C)
        q, k, v = self.q(h), self.k(h), self.v(h)
        outs = []
        for i in range(0, q.shape[1], self.chunk):
            s = torch.matmul(q[:, i:i + self.chunk], k.transpose(1, 2)) * self.scale
            p = torch.softmax(s.float(), dim=-1).to(v.dtype)
            outs.append(torch.matmul(p, v))
        o = self.out(torch.cat(outs, dim=1))
        return x + o.reshape(B, H, W, C).permute(0, 3, 1, 2)


class VAEDecoder(nn.Module):
    def __init__(self, latent_channels=4, channels=(512, 512, 256, 128),
                 out_channels=3, resnets_per_level=3, scaling_factor=0.13025):
        super().__init__()
        self.scaling_factor = scaling_factor
        c0 = channels[0]
        self.conv_in = Conv3x3(latent_channels, c0)
        self.mid_res1 = VAEResnet(c0, c0)
        self.mid_attn = VAEMidAttention(c0)
        self.mid_res2 = VAEResnet(c0, c0)
        self.levels = nn.ModuleList()
        c_prev = c0
        for li, c in enumerate(channels):
            blocks = nn.ModuleList()
            for _ in range(resnets_per_level):
                blocks.append(VAEResnet(c_prev, c))
                c_prev = c
            up = Conv3x3(c, c) if li < len(channels) - 1 else None
            self.levels.append(nn.ModuleList([blocks, nn.ModuleList([up] if up else [])]))
        self.norm_out = GroupNormSiLU(channels[-1])
        self.conv_out = Conv3x3(channels[-1], out_channels)

    def forward(self, z):
        h = self.conv_in(z / self.scaling_factor)
        h = self.mid_res2(self.mid_attn(self.mid_res1(h)))
        for blocks, ups in self.levels:
            for blk in blocks:
                h = blk(h)
            if len(ups):
                h = ups[0](h, upsample=True)  # fused nearest-2x + conv
        return self.conv_out(self.norm_out(h))


class VAEDecoderSmall(VAEDecoder):
    def __init__(self):
        super().__init__(channels=(64, 64, 32, 32))
