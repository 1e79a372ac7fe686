"""Public op API: gfx950 HIP kernels on GPU, fp32 torch references on CPU.

Dispatch contract: on a GPU tensor the hand-written kernel MUST run — if the
in-tree extension is missing the call raises rather than silently falling back
(build contract: "make your ops fail loudly if their extension is missing on a
GPU box").  CPU tensors use the fp32 reference path so the whole model stack is
testable without hardware.
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from . import reference as ref
from ._build import get_ext


def _ext_for(t: torch.Tensor, *grad_tensors, any_dtype: bool = False):
    """Dispatch: hand kernel on GPU for inference; differentiable torch
    composition when autograd needs to flow (training forward) — the custom
    kernels are forward-only, so grad-mode falls back to reference math on
    the SAME device (hipBLASLt/eager ROCm kernels), keeping LoRA/backward
    correct end-to-end.

    Most kernels are bf16-native (CDNA4 MFMA tiles); a non-bf16 GPU tensor
    (e.g. an fp32 research model) runs the reference composition on the same
    device rather than erroring.  Ops whose kernels take fp32 (sampling,
    softmax, AdamW) pass any_dtype=True.  A MISSING extension on a GPU box
    still fails loudly (get_ext(required=True))."""
    if torch.is_grad_enabled() and any(
            isinstance(g, torch.Tensor) and g.requires_grad
            for g in (t, *grad_tensors)):
        return None
    if t.is_cuda:
        if not any_dtype and t.dtype is not torch.bfloat16:
            get_ext(required=True)  # still verify the .so is present
            return None
        ext = get_ext(required=True)
        from ..gpu.guard import SyncProxy, debug_sync_enabled

        return SyncProxy(ext) if debug_sync_enabled() else ext
    return None


def attention(q, k, v, causal: bool = False, scale: Optional[float] = None):
    """Flash attention fwd (K1/K5/K7). q [B,Hq,Sq,D], k/v [B,Hkv,Sk,D] bf16."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    ext = _ext_for(q, k, v)
    if ext is None:
        return ref.attention_ref(q, k, v, causal, scale)
    return ext.attention(q.contiguous(), k.contiguous(), v.contiguous(), causal, scale)


def attention_qkv(q, k, v, causal: bool = False, scale: Optional[float] = None):
    """Transpose-free attention for sequence-major models.

    q: [B, S, Hq, D] (any strides, d contiguous — e.g. a slice of a fused QKV
    projection); k/v: [B, Sk, Hkv, D].  Returns [B, S, Hq*D] contiguous,
    ready for the output projection.  On GPU the kernel reads the strided
    layouts directly and writes BSHD — no .contiguous() copies at all.
    """
    B, S, Hq, D = q.shape
    Sk, Hkv = k.shape[1], k.shape[2]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    ext = _ext_for(q, k, v)
    if ext is None:
        o = ref.attention_ref(q.transpose(1, 2), k.transpose(1, 2),
                              v.transpose(1, 2), causal, scale)
        return o.transpose(1, 2).reshape(B, S, Hq * D)
    o = ext.attention_bshd(q.transpose(1, 2), k.transpose(1, 2),
                           v.transpose(1, 2), causal, scale)
    return o.view(B, S, Hq * D)


def paged_decode(q, k_cache, v_cache, block_table, seq_lens, block_size: int = 0,
                 scale: Optional[float] = None):
    """Decode attention over a (paged) KV cache (K6). q [B,Hq,D].
    Caches may be bf16 or float8_e4m3fn (fp8 halves KV bytes/token)."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    ext = _ext_for(q)
    if ext is None:
        if k_cache.dtype not in (torch.float32, torch.bfloat16, torch.float16):
            k_cache = k_cache.to(torch.float32)
            v_cache = v_cache.to(torch.float32)
        return ref.paged_decode_ref(q, k_cache, v_cache, block_table, seq_lens,
                                    block_size, scale)
    # .contiguous() also materializes expand()ed (stride-0) block tables —
    # the kernel indexes bt + row*stride and would read out of bounds.
    # block_table=None = contiguous (non-paged) cache mode.
    bt = block_table.contiguous() if block_table is not None else None
    return ext.paged_decode(q.contiguous(), k_cache, v_cache, bt,
                            seq_lens.int(), block_size, scale)


def groupnorm_silu(x, gamma, beta, groups: int = 32, eps: float = 1e-5,
                   do_silu: bool = True):
    ext = _ext_for(x, gamma, beta)
    if ext is None:
        return ref.groupnorm_silu_ref(x, gamma, beta, groups, eps, do_silu)
    return ext.groupnorm_silu(x.contiguous(), gamma.float().contiguous(),
                              beta.float().contiguous(), groups, eps, do_silu)


def layernorm(x, gamma, beta, eps: float = 1e-5):
    ext = _ext_for(x, gamma, beta)
    if ext is None:
        return ref.layernorm_ref(x, gamma, beta, eps)
    return ext.layernorm(x.contiguous(), gamma.float().contiguous(),
                         beta.float().contiguous(), eps)


def rmsnorm(x, gamma, eps: float = 1e-6):
    ext = _ext_for(x, gamma)
    if ext is None:
        return ref.rmsnorm_ref(x, gamma, eps)
    return ext.rmsnorm(x.contiguous(), gamma.float().contiguous(), eps)


def cfg_euler(x_t, eps_c, eps_u, guidance: float, dsigma: float):
    """Fused CFG combine + Euler update (K4)."""
    ext = _ext_for(x_t)
    if ext is None:
        return ref.cfg_euler_ref(x_t, eps_c, eps_u, guidance, dsigma)
    return ext.cfg_euler(x_t.contiguous(), eps_c.contiguous(),
                         eps_u.contiguous() if eps_u is not None else None,
                         guidance, dsigma)


def silu_mul(a, b):
    ext = _ext_for(a, b)
    if ext is None:
        return ref.silu_mul_ref(a, b)
    return ext.silu_mul(a.contiguous(), b.contiguous())


def geglu(a, b):
    ext = _ext_for(a, b)
    if ext is None:
        return ref.geglu_ref(a, b)
    return ext.geglu(a.contiguous(), b.contiguous())


def glu_fused(src, gelu: bool = False):
    """act(src[..., :I]) * src[..., I:] reading the fused projection output
    in place (no slice copies).  gelu=True -> GEGLU (SDXL FF), else SwiGLU
    (Llama gate_up)."""
    inner = src.shape[-1] // 2
    ext = _ext_for(src)
    if ext is None or inner % 8 != 0:
        a, b = src[..., :inner], src[..., inner:]
        return (ref.geglu_ref(a, b) if gelu else ref.silu_mul_ref(a, b))
    return ext.glu_fused(src.contiguous(), gelu)


def add_residual(a, b):
    ext = _ext_for(a, b)
    if ext is None:
        return (a.float() + b.float()).to(a.dtype)
    return ext.add_residual(a.contiguous(), b.contiguous())


def rope(x, cos, sin, positions=None, inplace: bool = False):
    """RoPE with host-precomputed tables. x [B,H,S,D] bf16.
    positions: None (0..S-1), [S] (shared across batch) or [B*S] (per-seq —
    the batched-decode case where every sequence sits at its own offset)."""
    ext = _ext_for(x)
    if ext is None:
        if positions is not None and positions.numel() == x.shape[0] * x.shape[2] \
                and x.shape[0] > 1:
            B, H, S, D = x.shape
            pos = positions.view(B, S).long()
            outs = [ref.rope_ref(x[b:b + 1], cos, sin, pos[b]) for b in range(B)]
            return torch.cat(outs, 0)
        return ref.rope_ref(x, cos, sin, positions.long() if positions is not None else None)
    # kernel is stride-aware (d contiguous required); inplace works on views
    # (e.g. q/k slices of a fused QKV projection — no copies at all)
    assert x.stride(-1) == 1, "rope needs contiguous head_dim"
    y = x if inplace else x.clone()
    pos = positions
    if pos is not None:
        pos = pos.int()
        if pos.numel() == x.shape[2] and x.shape[0] > 1:
            pos = pos.repeat(x.shape[0])
    ext.rope_(y, cos.float().contiguous(), sin.float().contiguous(), pos)
    return y


def rope_tables(max_seq: int, dim: int, base: float = 10000.0, device="cpu"):
    """Host-side cos/sin tables (guide App.B: no on-device trig)."""
    inv = 1.0 / (base ** (torch.arange(0, dim, 2, dtype=torch.float64) / dim))
    t = torch.arange(max_seq, dtype=torch.float64)
    freqs = torch.outer(t, inv)
    return (freqs.cos().float().to(device), freqs.sin().float().to(device))


def repack_conv3x3_weight(weight: torch.Tensor) -> torch.Tensor:
    """[K,C,3,3] conv weight → the kernel's [9, Kpad, C16] layout (c
    contiguous per (tap, k) row so the MFMA A-fragment is one bf16x8 read;
    K padded to the 64-channel block, C to the 16-channel chunk)."""
    K, C = weight.shape[0], weight.shape[1]
    kpad = (K + 63) // 64 * 64
    c16 = (C + 15) // 16 * 16
    wr = torch.zeros(9, kpad, c16, dtype=torch.bfloat16, device=weight.device)
    w = weight.permute(2, 3, 0, 1).reshape(9, K, C)  # [tap, k, c]
    wr[:, :K, :C] = w.to(torch.bfloat16)
    return wr.contiguous()


def conv3x3(x, wr, bias, K: int, residual=None, raw_weight=None,
            upsample: bool = False):
    """3x3 stride-1 pad-1 conv (K3). x [N,C,H,W] bf16; wr from
    repack_conv3x3_weight; bias fp32 [K]; optional fused residual add;
    upsample=True fuses a nearest-2x upsample of x into the conv read.
    raw_weight [K,C,3,3] drives the CPU/autograd reference path."""
    ext = _ext_for(x, raw_weight if raw_weight is not None else x)
    if ext is None:
        assert raw_weight is not None, "reference conv path needs raw_weight"
        xin = x
        if upsample:
            xin = torch.nn.functional.interpolate(x, scale_factor=2.0,
                                                  mode="nearest")
        y = torch.nn.functional.conv2d(
            xin.float(), raw_weight.float(), bias.float(), padding=1)
        if residual is not None:
            y = y + residual.float()
        return y.to(x.dtype)
    out = ext.conv3x3(x.contiguous(), wr, bias,
                      residual.contiguous() if residual is not None else None,
                      K, upsample)
    return out


def conv3x3_gn(x, wr, bias, K: int, gamma, beta, groups: int = 32,
               eps: float = 1e-5, residual=None, raw_weight=None,
               upsample: bool = False):
    """GroupNorm+SiLU fused into the K3 conv's staging read: one stats pass
    over x + the conv — the gn_norm write/read pass over the full tensor
    disappears."""
    ext = _ext_for(x, raw_weight if raw_weight is not None else x)
    if ext is None:
        assert raw_weight is not None
        h = torch.nn.functional.group_norm(x.float(), groups, gamma.float(),
                                           beta.float(), eps)
        h = torch.nn.functional.silu(h)
        if upsample:
            h = torch.nn.functional.interpolate(h, scale_factor=2.0,
                                                mode="nearest")
        y = torch.nn.functional.conv2d(h, raw_weight.float(), bias.float(),
                                       padding=1)
        if residual is not None:
            y = y + residual.float()
        return y.to(x.dtype)
    return ext.conv3x3_gn(x.contiguous(), wr, bias,
                          residual.contiguous() if residual is not None else None,
                          K, upsample, gamma.float().contiguous(),
                          beta.float().contiguous(), groups, eps)


def adamw_step(p, g, m, v, lr, beta1=0.9, beta2=0.999, eps=1e-8, wd=0.01, step=1):
    """Fused AdamW (K9); in-place on p/m/v."""
    ext = _ext_for(p, any_dtype=True)
    if ext is None:
        return ref.adamw_ref(p, g, m, v, lr, beta1, beta2, eps, wd, step)
    ext.adamw_(p, g.contiguous(), m, v, lr, beta1, beta2, eps, wd, step)
    return p


def sample(logits, temperature: float = 1.0, seed: int = 0):
    """Fused sampling (K8): gumbel-max over softmax(logits/T); T=0 → argmax."""
    ext = _ext_for(logits, any_dtype=True)
    if ext is None:
        if temperature <= 0:
            return logits.argmax(-1).int()
        g = torch.Generator(device="cpu").manual_seed(seed or 1)
        u = torch.rand(logits.shape, generator=g).clamp_min(1e-10)
        gumbel = -(-u.log()).log()
        return (logits.float() / temperature + gumbel).argmax(-1).int()
    return ext.sample_gumbel(logits.float().contiguous(), temperature, seed)


def softmax(x):
    ext = _ext_for(x, any_dtype=True)
    if ext is None:
        return torch.softmax(x.float(), dim=-1)
    return ext.softmax_fwd(x.float().contiguous())
