"""Plain PyTorch fp32 reference implementations of every HIP op.

These are the ground truth the GPU numerics tests compare the gfx950 kernels
against (per the build contract: "numerics tests for a HIP kernel compare it
against a plain PyTorch fp32 reference of the same op"), and the CPU execution
path for hermetic tests.
"""
from __future__ import annotations

import math
from typing import Optional

import torch


def attention_ref(q, k, v, causal: bool = False, scale: Optional[float] = None):
    """q [B,Hq,Sq,D], k/v [B,Hkv,Sk,D] → [B,Hq,Sq,D]; GQA by head repeat."""
    B, Hq, Sq, D = q.shape
    Hkv, Sk = k.shape[1], k.shape[2]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    qf, kf, vf = q.float(), k.float(), v.float()
    if Hkv != Hq:
        rep = Hq // Hkv
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        off = Sk - Sq
        mask = torch.ones(Sq, Sk, dtype=torch.bool, device=q.device).tril(off)
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, vf).to(q.dtype)


def paged_decode_ref(q, k_cache, v_cache, block_table, seq_lens, block_size,
                     scale: Optional[float] = None):
    """q [B,Hq,D]; paged cache [nblocks,Hkv,block_size,D] (+table) or
    contiguous [B,Hkv,S,D] when block_table is None."""
    B, Hq, D = q.shape
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    Hkv = k_cache.shape[1]
    G = Hq // Hkv
    out = torch.empty_like(q)
    for b in range(B):
        S = int(seq_lens[b])
        if block_table is None:
            kb = k_cache[b, :, :S]  # [Hkv,S,D]
            vb = v_cache[b, :, :S]
        else:
            nblk = (S + block_size - 1) // block_size
            ks, vs = [], []
            for i in range(nblk):
                blk = int(block_table[b, i])
                ks.append(k_cache[blk])  # [Hkv, block_size, D]
                vs.append(v_cache[blk])
            kb = torch.cat(ks, dim=1)[:, :S]
            vb = torch.cat(vs, dim=1)[:, :S]
        for h in range(Hq):
            hk = h // G
            s = (kb[hk].float() @ q[b, h].float()) * scale  # [S]
            p = torch.softmax(s, dim=-1)
            out[b, h] = (p @ vb[hk].float()).to(q.dtype)
    return out


def groupnorm_silu_ref(x, gamma, beta, groups, eps=1e-5, do_silu=True):
    y = torch.nn.functional.group_norm(
        x.float(), groups, gamma.float(), beta.float(), eps
    )
    if do_silu:
        y = torch.nn.functional.silu(y)
    return y.to(x.dtype)


def layernorm_ref(x, gamma, beta, eps=1e-5):
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), gamma.float(), beta.float(), eps
    ).to(x.dtype)


def rmsnorm_ref(x, gamma, eps=1e-6):
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd * gamma.float()).to(x.dtype)


def cfg_euler_ref(x_t, eps_c, eps_u, guidance, dsigma):
    e = eps_c.float()
    if eps_u is not None:
        e = eps_u.float() + guidance * (eps_c.float() - eps_u.float())
    return (x_t.float() + dsigma * e).to(x_t.dtype)


def silu_mul_ref(a, b):
    return (torch.nn.functional.silu(a.float()) * b.float()).to(a.dtype)


def geglu_ref(a, b):
    return (torch.nn.functional.gelu(a.float(), approximate="tanh") * b.float()).to(a.dtype)


def rope_ref(x, cos, sin, positions=None):
    """x [B,H,S,D]; cos/sin [S_max, D/2] f32; rotate-half convention."""
    B, H, S, D = x.shape
    idx = positions if positions is not None else torch.arange(S, device=x.device)
    c = cos[idx].view(1, 1, S, D // 2).float()
    s = sin[idx].view(1, 1, S, D // 2).float()
    x1 = x[..., : D // 2].float()
    x2 = x[..., D // 2:].float()
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    return out.to(x.dtype)


def adamw_ref(p, g, m, v, lr, beta1, beta2, eps, wd, step):
    pf, gf = p.float(), g.float()
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    mhat = m / (1 - beta1**step)
    vhat = v / (1 - beta2**step)
    pf = pf - lr * (mhat / (vhat.sqrt() + eps) + wd * pf)
    p.copy_(pf.to(p.dtype))
    return p
