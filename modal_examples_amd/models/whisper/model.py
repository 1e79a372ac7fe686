"""Whisper-large-v3-class encoder/decoder, MI355X-first.

The model behind the reference's canonical batch-transcription example
(trigger: 06_gpu_and_ml/speech-to-text/batched_whisper.py:93-138 —
Whisper-large-v3 via transformers pipeline; streaming_whisper.py:139).

MI355X mapping: encoder bidirectional attention
(S=1500, 20 heads × D64) → gfx950 flash kernel (non-causal); decoder causal
self-attention + cross-attention over a contiguous KV cache → gfx950 decode
kernel; LayerNorm/GELU fusions → gfx950 kernels; conv1d frontend → MIOpen.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn

from ...ops import functional as OF
from ..sdxl.layers import LayerNormK


@dataclass
class WhisperConfig:
    n_mels: int = 128
    n_audio_ctx: int = 1500
    n_state: int = 1280
    n_head: int = 20
    n_audio_layer: int = 32
    n_text_ctx: int = 448
    n_text_layer: int = 32
    vocab_size: int = 51866

    @staticmethod
    def large_v3() -> "WhisperConfig":
        return WhisperConfig()

    @staticmethod
    def small_test() -> "WhisperConfig":
        return WhisperConfig(n_mels=80, n_audio_ctx=100, n_state=128, n_head=2,
                             n_audio_layer=2, n_text_ctx=64, n_text_layer=2,
                             vocab_size=512)


def sinusoids(length: int, channels: int) -> torch.Tensor:
    log_timescale = math.log(10000.0) / (channels // 2 - 1)
    inv = torch.exp(-log_timescale * torch.arange(channels // 2, dtype=torch.float32))
    scaled = torch.arange(length, dtype=torch.float32)[:, None] * inv[None]
    return torch.cat([scaled.sin(), scaled.cos()], dim=1)


class MHA(nn.Module):
    """Self/cross attention with fused QKV (self) or fused KV (cross)."""

    def __init__(self, state: int, heads: int):
        super().__init__()
        self.h = heads
        self.d = state // heads
        self.q = nn.Linear(state, state)
        self.kv = nn.Linear(state, 2 * state, bias=False)
        self.out = nn.Linear(state, state)

    def qkv(self, x, ctx=None):
        """BSHD views: q [B,S,h,d]; k/v [B,Sk,h,d] (no copies)."""
        B, S, C = x.shape
        src = ctx if ctx is not None else x
        Sk = src.shape[1]
        q = self.q(x).view(B, S, self.h, self.d)
        kv = self.kv(src).view(B, Sk, 2, self.h, self.d)
        return q, kv[:, :, 0], kv[:, :, 1]

    def forward(self, x, ctx=None, causal=False):
        B, S, C = x.shape
        q, k, v = self.qkv(x, ctx)
        return self.out(OF.attention_qkv(q, k, v, causal=causal))


class MLP(nn.Module):
    def __init__(self, state: int):
        super().__init__()
        self.fc1 = nn.Linear(state, 4 * state)
        self.fc2 = nn.Linear(4 * state, state)

    def forward(self, x):
        # gelu(a)*1 via the fused GEGLU kernel with ones is wasteful; plain gelu
        h = self.fc1(x)
        return self.fc2(torch.nn.functional.gelu(h, approximate="tanh"))


class EncoderLayer(nn.Module):
    def __init__(self, cfg: WhisperConfig):
        super().__init__()
        self.ln1 = LayerNormK(cfg.n_state)
        self.attn = MHA(cfg.n_state, cfg.n_head)
        self.ln2 = LayerNormK(cfg.n_state)
        self.mlp = MLP(cfg.n_state)

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        return x + self.mlp(self.ln2(x))


class DecoderLayer(nn.Module):
    def __init__(self, cfg: WhisperConfig):
        super().__init__()
        self.ln1 = LayerNormK(cfg.n_state)
        self.self_attn = MHA(cfg.n_state, cfg.n_head)
        self.ln2 = LayerNormK(cfg.n_state)
        self.cross_attn = MHA(cfg.n_state, cfg.n_head)
        self.ln3 = LayerNormK(cfg.n_state)
        self.mlp = MLP(cfg.n_state)


class WhisperModel(nn.Module):
    def __init__(self, cfg: WhisperConfig = None):
        super().__init__()
        cfg = cfg or WhisperConfig.large_v3()
        self.cfg = cfg
        s = cfg.n_state
        self.conv1 = nn.Conv1d(cfg.n_mels, s, 3, padding=1)
        self.conv2 = nn.Conv1d(s, s, 3, stride=2, padding=1)
        self.register_buffer("pos_audio", sinusoids(cfg.n_audio_ctx, s), persistent=False)
        self.enc_layers = nn.ModuleList(
            [EncoderLayer(cfg) for _ in range(cfg.n_audio_layer)])
        self.enc_ln = LayerNormK(s)

        self.tok_embed = nn.Embedding(cfg.vocab_size, s)
        self.pos_embed = nn.Parameter(torch.zeros(cfg.n_text_ctx, s))
        self.dec_layers = nn.ModuleList(
            [DecoderLayer(cfg) for _ in range(cfg.n_text_layer)])
        self.dec_ln = LayerNormK(s)

    # ------------------------------------------------ encoder (K5)

    @torch.no_grad()
    def encode(self, mel: torch.Tensor) -> torch.Tensor:
        """mel [B, n_mels, 2*n_audio_ctx] → audio features [B, n_audio_ctx, state]."""
        x = torch.nn.functional.gelu(self.conv1(mel))
        x = torch.nn.functional.gelu(self.conv2(x))
        x = x.permute(0, 2, 1)  # [B, T, state]
        x = x + self.pos_audio[: x.shape[1]].to(x.dtype)
        for layer in self.enc_layers:
            x = layer(x)
        return self.enc_ln(x)

    # ------------------------------------------------ decoder

    def forward_train(self, mel: torch.Tensor, tokens: torch.Tensor) -> torch.Tensor:
        """Teacher-forced training forward for ASR fine-tuning: grad-enabled,
        full-sequence causal decoder over encoded audio.  Returns logits
        [B, S, vocab] (fp32).  Attention dispatches to differentiable refs
        when grads are required (ops/functional grad-aware dispatch)."""
        x = torch.nn.functional.gelu(self.conv1(mel))
        x = torch.nn.functional.gelu(self.conv2(x))
        x = x.permute(0, 2, 1)
        x = x + self.pos_audio[: x.shape[1]].to(x.dtype)
        for layer in self.enc_layers:
            x = layer(x)
        audio = self.enc_ln(x)

        B, S = tokens.shape
        y = self.tok_embed(tokens) + self.pos_embed[:S].to(audio.dtype)
        for layer in self.dec_layers:
            y = y + layer.self_attn(layer.ln1(y), causal=True)
            y = y + layer.cross_attn(layer.ln2(y), ctx=audio)
            y = y + layer.mlp(layer.ln3(y))
        y = self.dec_ln(y)
        return (y @ self.tok_embed.weight.T.to(y.dtype)).float()

    @torch.no_grad()
    def decode_prefill(self, tokens: torch.Tensor, audio: torch.Tensor, caches):
        """tokens [B,S]; audio [B,T,state]; caches: per-layer dict with
        k/v [B,H,ctx,D] self cache + cross k/v computed here."""
        B, S = tokens.shape
        x = self.tok_embed(tokens) + self.pos_embed[:S].to(audio.dtype)
        for li, layer in enumerate(self.dec_layers):
            c = caches[li]
            h = layer.ln1(x)
            q, k, v = layer.self_attn.qkv(h)  # BSHD
            c["k"][:, :, :S] = k.transpose(1, 2)
            c["v"][:, :, :S] = v.transpose(1, 2)
            x = x + layer.self_attn.out(OF.attention_qkv(q, k, v, causal=True))
            if "ck" not in c:
                _, ck, cv = layer.cross_attn.qkv(h, audio)  # BSHD
                # decode kernel wants [B, H, T, D] contiguous caches
                c["ck"] = ck.transpose(1, 2).contiguous()
                c["cv"] = cv.transpose(1, 2).contiguous()
            elif c.get("persistent"):
                # graph-captured decode: cross caches are STATIC buffers
                # refreshed in place per call
                _, ck, cv = layer.cross_attn.qkv(h, audio)
                c["ck"].copy_(ck.transpose(1, 2))
                c["cv"].copy_(cv.transpose(1, 2))
            x = x + layer.cross_attn(layer.ln2(x), ctx=audio)
            x = x + layer.mlp(layer.ln3(x))
        x = self.dec_ln(x[:, -1:])
        return (x @ self.tok_embed.weight.T.to(x.dtype))[:, 0].float()

    @torch.no_grad()
    def decode_step_dev(self, tokens: torch.Tensor, pos_d: torch.Tensor,
                        caches, audio_len: int):
        """Graph-capturable single-token step: the position lives ON DEVICE
        (int64 [1]) so the whole step replays as one hipGraph (K10's role for
        the Whisper decoder, like the Llama engine's captured decode)."""
        B = tokens.shape[0]
        x = self.tok_embed(tokens).unsqueeze(1) + self.pos_embed.index_select(
            0, pos_d).to(self.tok_embed.weight.dtype)
        lens_self = (pos_d + 1).to(torch.int32).expand(B).contiguous()
        lens_cross = torch.full((B,), audio_len, dtype=torch.int32,
                                device=x.device)
        for li, layer in enumerate(self.dec_layers):
            c = caches[li]
            h = layer.ln1(x)
            q, k, v = layer.self_attn.qkv(h)  # BSHD [B,1,h,d]
            c["k"].index_copy_(2, pos_d, k.transpose(1, 2))
            c["v"].index_copy_(2, pos_d, v.transpose(1, 2))
            o = OF.paged_decode(q[:, 0].contiguous(), c["k"], c["v"], None,
                                lens_self)
            x = x + layer.self_attn.out(o.reshape(B, 1, -1))
            h2 = layer.ln2(x)
            q2 = layer.cross_attn.q(h2).view(B, 1, -1)
            q2 = q2.view(B, layer.cross_attn.h, layer.cross_attn.d)
            o2 = OF.paged_decode(q2.contiguous(), c["ck"], c["cv"], None,
                                 lens_cross)
            x = x + layer.cross_attn.out(o2.reshape(B, 1, -1))
            x = x + layer.mlp(layer.ln3(x))
        x = self.dec_ln(x)
        return (x @ self.tok_embed.weight.T.to(x.dtype))[:, 0].float()

    def decode_step(self, tokens: torch.Tensor, pos: int, caches, audio_len: int):
        """Single-token step for all sequences; contiguous caches (K6 kernel)."""
        B = tokens.shape[0]
        x = self.tok_embed(tokens).unsqueeze(1) + self.pos_embed[pos:pos + 1].to(
            self.tok_embed.weight.dtype)
        lens_self = torch.full((B,), pos + 1, dtype=torch.int32, device=x.device)
        lens_cross = torch.full((B,), audio_len, dtype=torch.int32, device=x.device)
        for li, layer in enumerate(self.dec_layers):
            c = caches[li]
            h = layer.ln1(x)
            q, k, v = layer.self_attn.qkv(h)  # BSHD [B,1,h,d]
            c["k"][:, :, pos:pos + 1] = k.transpose(1, 2)
            c["v"][:, :, pos:pos + 1] = v.transpose(1, 2)
            o = OF.paged_decode(q[:, 0].contiguous(), c["k"], c["v"], None, lens_self)
            x = x + layer.self_attn.out(o.reshape(B, 1, -1))
            h2 = layer.ln2(x)
            q2 = layer.cross_attn.q(h2).view(B, 1, -1)
            q2 = q2.view(B, layer.cross_attn.h, layer.cross_attn.d)
            o2 = OF.paged_decode(q2.contiguous(), c["ck"], c["cv"], None, lens_cross)
            x = x + layer.cross_attn.out(o2.reshape(B, 1, -1))
            x = x + layer.mlp(layer.ln3(x))
        x = self.dec_ln(x)
        return (x @ self.tok_embed.weight.T.to(x.dtype))[:, 0].float()
