"""Llama-3-class decoder-only transformer, MI355X-first.

The model family behind the reference's canonical LLM-serving examples
(trigger: 06_gpu_and_ml/llm-serving/vllm_inference.py:158-209 — Llama/Qwen
class models under vLLM; trtllm_latency.py:10-21 LLaMA-3-8B).

MI355X mapping:
  K7 prefill  → gfx950 flash attention (causal, GQA, D=128)
  K6 decode   → gfx950 paged decode kernel over the paged KV cache
  RMSNorm / RoPE / SwiGLU → fused gfx950 kernels
  QKV and gate_up projections → single fused GEMMs (hipBLASLt)
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from ...ops import functional as OF
from ..sdxl.layers import RMSNormK


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    dim: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    head_dim: int = 128
    ffn_dim: int = 14336
    rope_base: float = 500000.0
    norm_eps: float = 1e-5
    max_seq: int = 8192
    n_experts: int = 0   # >0: MoE FFN (Mixtral/DeepSeek-class)
    experts_per_tok: int = 2

    @staticmethod
    def llama3_8b() -> "LlamaConfig":
        return LlamaConfig()

    @staticmethod
    def small() -> "LlamaConfig":
        return LlamaConfig(vocab_size=1024, dim=256, n_layers=2, n_heads=4,
                           n_kv_heads=2, head_dim=64, ffn_dim=512,
                           rope_base=10000.0, max_seq=512)

    @staticmethod
    def moe_small() -> "LlamaConfig":
        return LlamaConfig(vocab_size=1024, dim=256, n_layers=2, n_heads=4,
                           n_kv_heads=2, head_dim=64, ffn_dim=256,
                           rope_base=10000.0, max_seq=512, n_experts=4)


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_world: int = 1):
        """tp_world>1: Megatron head-sharded block (the --tensor-parallel-size
        role of vllm_inference.py:180 / trtllm tensor_parallel_size).  qkv and
        gate_up are column-sharded (this rank's q/kv-head and gate/up slices,
        NO comm in forward), o_proj and down are row-sharded — the model adds
        ONE all-reduce after each (two collectives per block over xGMI)."""
        super().__init__()
        d, hd = cfg.dim, cfg.head_dim
        assert cfg.n_heads % tp_world == 0 and cfg.n_kv_heads % tp_world == 0 \
            and cfg.ffn_dim % tp_world == 0, \
            f"tp={tp_world} must divide heads {cfg.n_heads}/{cfg.n_kv_heads}" \
            f" and ffn {cfg.ffn_dim}"
        self.nq, self.nkv = cfg.n_heads // tp_world, cfg.n_kv_heads // tp_world
        self.attn_norm = RMSNormK(d, cfg.norm_eps)
        self.qkv = nn.Linear(d, (self.nq + 2 * self.nkv) * hd, bias=False)
        self.o_proj = nn.Linear(self.nq * hd, d, bias=False)
        self.ffn_norm = RMSNormK(d, cfg.norm_eps)
        if cfg.n_experts > 0:
            assert tp_world == 1, "MoE + TP not supported (EP is a non-goal)"
            self.moe = MoEFFN(d, cfg.ffn_dim, cfg.n_experts,
                              cfg.experts_per_tok)
            self.gate_up = self.down = None
        else:
            self.moe = None
            self.gate_up = nn.Linear(d, 2 * cfg.ffn_dim // tp_world,
                                     bias=False)
            self.down = nn.Linear(cfg.ffn_dim // tp_world, d, bias=False)
        self.hd = hd
        self.ffn_dim = cfg.ffn_dim // tp_world

    def project_qkv(self, x):
        """x [B,S,d] → BSHD views of the fused projection (zero copies):
        q [B,S,nq,hd], k/v [B,S,nkv,hd]."""
        B, S, _ = x.shape
        qkv = self.qkv(x)
        q, k, v = qkv.split(
            [self.nq * self.hd, self.nkv * self.hd, self.nkv * self.hd], dim=-1
        )
        return (q.view(B, S, self.nq, self.hd), k.view(B, S, self.nkv, self.hd),
                v.view(B, S, self.nkv, self.hd))

    def ffn(self, x):
        if self.moe is not None:
            return self.moe(x)
        # fused kernel reads both halves of the gate_up output in place
        return self.down(OF.glu_fused(self.gate_up(x), gelu=False))


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp=None):
        """tp: optional parallel.tp.TPGroup — shards every block across the
        group (embed/norms/lm_head replicated; activations stay replicated,
        so logits are identical on every rank and sampling needs no extra
        broadcast)."""
        super().__init__()
        self.cfg = cfg
        self.tp = tp if (tp is not None and tp.world > 1) else None
        w = self.tp.world if self.tp else 1
        self.embed = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.blocks = nn.ModuleList(
            [LlamaBlock(cfg, tp_world=w) for _ in range(cfg.n_layers)])
        self.norm = RMSNormK(cfg.dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
        # NOT buffers: module .to(bf16) must not downcast the f32 trig tables
        # (forced onto cpu so meta-device construction still gets real tables)
        with torch.device("cpu"):
            cos, sin = OF.rope_tables(cfg.max_seq, cfg.head_dim, cfg.rope_base)
        self._rope_cpu = (cos, sin)
        self._rope_cache = {}

    def _ar(self, t):
        """Sum-reduce a row-parallel partial across the TP group (no-op
        single-rank)."""
        return self.tp.all_reduce(t) if self.tp is not None else t

    def _rope_tables(self, device) -> tuple:
        key = str(device)
        if key not in self._rope_cache:
            self._rope_cache[key] = tuple(t.to(device) for t in self._rope_cpu)
        return self._rope_cache[key]

    @torch.no_grad()
    def prefill(self, tokens: torch.Tensor, kv_writer=None, last_pos=None):
        """tokens [B,S] → logits [B, vocab] (last position only).
        kv_writer(layer_idx, k, v): callback storing [B,nkv,S,hd] into cache.
        last_pos [B] int64: per-row index of the last REAL token — the ragged
        path right-pads shorter prompts and samples at each row's own end
        (vLLM-style mixed-length prefill; role of vllm_inference.py:158-209)."""
        x = self.embed(tokens)
        rc, rs = self._rope_tables(x.device)
        gpu = x.is_cuda
        for li, blk in enumerate(self.blocks):
            h = blk.attn_norm(x)
            q, k, v = blk.project_qkv(h)  # BSHD views
            q = OF.rope(q.transpose(1, 2), rc, rs, inplace=gpu).transpose(1, 2)
            k = OF.rope(k.transpose(1, 2), rc, rs, inplace=gpu).transpose(1, 2)
            if kv_writer is not None:
                kv_writer(li, k, v)  # [B,S,nkv_local,hd]
            x = x + self._ar(blk.o_proj(OF.attention_qkv(q, k, v, causal=True)))
            x = x + self._ar(blk.ffn(blk.ffn_norm(x)))
        if last_pos is None:
            x = self.norm(x[:, -1:])
        else:
            idx = last_pos.view(-1, 1, 1).expand(-1, 1, x.shape[-1])
            x = self.norm(x.gather(1, idx))
        return self.lm_head(x)[:, 0].float()

    @torch.no_grad()
    def decode_step(self, tokens, positions, kv_append, kv_attend):
        """One token per sequence. tokens [B], positions [B] int32;
        kv_append(li, k, v): store [B,nkv,1,hd] at per-seq positions;
        kv_attend(li, q): paged attention of q [B,nq,hd] vs the cache."""
        x = self.embed(tokens).unsqueeze(1)  # [B,1,d]
        rc, rs = self._rope_tables(x.device)
        gpu = x.is_cuda
        for li, blk in enumerate(self.blocks):
            h = blk.attn_norm(x)
            q, k, v = blk.project_qkv(h)  # BSHD views [B,1,h,hd]
            q = OF.rope(q.transpose(1, 2), rc, rs, positions=positions,
                        inplace=gpu).transpose(1, 2)
            k = OF.rope(k.transpose(1, 2), rc, rs, positions=positions,
                        inplace=gpu).transpose(1, 2)
            kv_append(li, k, v)  # [B,1,nkv_local,hd]
            o = kv_attend(li, q[:, 0].contiguous() if not q[:, 0].is_contiguous() else q[:, 0])
            x = x + self._ar(blk.o_proj(o.reshape(o.shape[0], 1, -1)))
            x = x + self._ar(blk.ffn(blk.ffn_norm(x)))
        x = self.norm(x[:, -1:])
        return self.lm_head(x)[:, 0].float()


def shard_llama_state(state: dict, cfg: LlamaConfig, rank: int,
                      world: int) -> dict:
    """Slice a FULL single-GPU state dict into rank's TP shard.

    Matches LlamaBlock(tp_world=world): per block, qkv/gate_up rows are this
    rank's q/kv-head and gate/up slices, o_proj/down columns are the
    matching input slices; embed/norms/lm_head replicate.  Used to TP-ify a
    baked checkpoint (fastload file) without materializing N copies.
    """
    hd = cfg.head_dim
    nq_l, nkv_l = cfg.n_heads // world, cfg.n_kv_heads // world
    ffn_l = cfg.ffn_dim // world
    q_off, k_off = 0, cfg.n_heads * hd
    v_off = k_off + cfg.n_kv_heads * hd
    out = {}
    for k, v in state.items():
        if k.endswith(".qkv.weight"):
            rows = torch.cat([
                v[q_off + rank * nq_l * hd: q_off + (rank + 1) * nq_l * hd],
                v[k_off + rank * nkv_l * hd: k_off + (rank + 1) * nkv_l * hd],
                v[v_off + rank * nkv_l * hd: v_off + (rank + 1) * nkv_l * hd],
            ])
            out[k] = rows
        elif k.endswith(".o_proj.weight"):
            out[k] = v[:, rank * nq_l * hd: (rank + 1) * nq_l * hd].contiguous()
        elif k.endswith(".gate_up.weight"):
            out[k] = torch.cat([
                v[rank * ffn_l: (rank + 1) * ffn_l],
                v[cfg.ffn_dim + rank * ffn_l: cfg.ffn_dim + (rank + 1) * ffn_l],
            ])
        elif k.endswith(".down.weight"):
            out[k] = v[:, rank * ffn_l: (rank + 1) * ffn_l].contiguous()
        else:
            out[k] = v
    return out


class MoEFFN(nn.Module):
    """Top-k mixture-of-experts FFN (the MoE model class the reference
    serves: deepseek_v4.py, gpt_oss_inference.py, misc/trtllm_deepseek.py —
    Mixtral/DeepSeek-style routed experts).

    Router scores per token; top-k experts run their SwiGLU on the tokens
    routed to them (token-dispatch loop: E is small, every expert GEMM is a
    dense hipBLASLt call over its token group; glu_fused reads both halves
    of the fused gate_up in place)."""

    def __init__(self, dim: int, ffn_dim: int, n_experts: int, top_k: int):
        super().__init__()
        self.router = nn.Linear(dim, n_experts, bias=False)
        self.gate_up = nn.Parameter(
            torch.empty(n_experts, 2 * ffn_dim, dim).normal_(std=0.02))
        self.down = nn.Parameter(
            torch.empty(n_experts, dim, ffn_dim).normal_(std=0.02))
        self.n_experts, self.top_k = n_experts, top_k

    def forward(self, x):
        from ...ops import functional as OF

        B, S, d = x.shape
        flat = x.reshape(-1, d)
        scores = self.router(flat).float()                  # [N, E]
        w, idx = scores.topk(self.top_k, dim=-1)            # [N, k]
        w = torch.softmax(w, dim=-1).to(x.dtype)
        out = torch.zeros_like(flat)
        for e in idx.unique().tolist():
            hit = (idx == e)                                # [N, k]
            rows = hit.any(-1).nonzero(as_tuple=True)[0]
            contrib = (w * hit).sum(-1)[rows].unsqueeze(-1)  # routed weight
            h = OF.glu_fused(flat[rows] @ self.gate_up[e].T, gelu=False)
            out[rows] += contrib * (h @ self.down[e].T)
        return out.view(B, S, d)
