"""Tensor parallelism over RCCL/xGMI for models beyond one GPU's memory.

Reference role: big-model TP serving (llm-serving's multi-GPU rows, e.g.
Llama-70B/405B split across a node).  MI355X sizing notes: 288 GB HBM3E per
GPU holds ~140B bf16 params, so TP is for the 70B-405B class — and xGMI is
point-to-point (7 links x ~153 GB/s, no switch), so every TP all-reduce is
per-link bound: prefer TP=2/4 with bigger shards over TP=8, and size
activations so the one reduce per block stays a small fraction of the GEMM.

Megatron-style pairing: ColumnParallelLinear (shards out_features; no comm
in forward) feeds RowParallelLinear (shards in_features; ONE all-reduce on
the output).  An attention or MLP block then costs exactly one collective.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn


class TPGroup:
    """A tensor-parallel process group (defaults to WORLD)."""

    def __init__(self, group=None):
        self.group = group
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.world > 1:
            if t.dtype is torch.bfloat16 and not t.is_cuda:
                # gloo (the CPU test backend) has no bf16 reduction; RCCL
                # reduces bf16 natively on GPU
                t32 = t.float()
                dist.all_reduce(t32, group=self.group)
                t.copy_(t32)
            else:
                dist.all_reduce(t, group=self.group)
        return t

    def all_gather_cat(self, t: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if self.world == 1:
            return t
        parts = [torch.empty_like(t) for _ in range(self.world)]
        dist.all_gather(parts, t.contiguous(), group=self.group)
        return torch.cat(parts, dim=dim)


class ColumnParallelLinear(nn.Module):
    """Shards out_features: each rank computes its slice of the output.
    No communication in forward (the following RowParallel layer reduces)."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 gather_output: bool = False, tp: Optional[TPGroup] = None,
                 dtype=None):
        super().__init__()
        self.tp = tp or TPGroup()
        assert out_features % self.tp.world == 0, \
            f"out_features {out_features} not divisible by tp={self.tp.world}"
        self.out_shard = out_features // self.tp.world
        self.linear = nn.Linear(in_features, self.out_shard, bias=bias,
                                dtype=dtype)
        self.gather_output = gather_output

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = self.linear(x)
        return self.tp.all_gather_cat(y) if self.gather_output else y


class RowParallelLinear(nn.Module):
    """Shards in_features: each rank holds a K-slice and produces a partial
    full-width output; ONE all-reduce sums the partials.  Bias is added
    after the reduce (once)."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 input_is_parallel: bool = True, tp: Optional[TPGroup] = None,
                 dtype=None):
        super().__init__()
        self.tp = tp or TPGroup()
        assert in_features % self.tp.world == 0, \
            f"in_features {in_features} not divisible by tp={self.tp.world}"
        self.in_shard = in_features // self.tp.world
        self.linear = nn.Linear(self.in_shard, out_features, bias=False,
                                dtype=dtype)
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None
        self.input_is_parallel = input_is_parallel

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.input_is_parallel:
            x = x.narrow(-1, self.tp.rank * self.in_shard, self.in_shard)
        y = self.linear(x)
        y = self.tp.all_reduce(y)
        if self.bias is not None:
            y = y + self.bias
        return y


def shard_linear(full: nn.Linear, kind: str, tp: Optional[TPGroup] = None):
    """Convert a full nn.Linear into this rank's parallel shard.

    kind='column': split rows of W (out_features).  kind='row': split
    columns of W (in_features).  Used to TP-ify an existing single-GPU
    checkpointed model without materializing per-rank weights twice.
    """
    tp = tp or TPGroup()
    out_f, in_f = full.weight.shape
    has_bias = full.bias is not None
    if kind == "column":
        m = ColumnParallelLinear(in_f, out_f, bias=has_bias, tp=tp,
                                 dtype=full.weight.dtype)
        rows = slice(tp.rank * m.out_shard, (tp.rank + 1) * m.out_shard)
        with torch.no_grad():
            m.linear.weight.copy_(full.weight[rows])
            if has_bias:
                m.linear.bias.copy_(full.bias[rows])
        return m
    if kind == "row":
        m = RowParallelLinear(in_f, out_f, bias=has_bias, tp=tp,
                              dtype=full.weight.dtype)
        cols = slice(tp.rank * m.in_shard, (tp.rank + 1) * m.in_shard)
        with torch.no_grad():
            m.linear.weight.copy_(full.weight[:, cols])
            if has_bias:
                m.bias.copy_(full.bias)
        return m
    raise ValueError(f"kind must be 'column' or 'row', got {kind!r}")


class TPMLP(nn.Module):
    """Megatron MLP: column (up, sharded activation) → row (down, one
    all-reduce).  The gelu runs on the shard — compute also divides by tp."""

    def __init__(self, d_model: int, d_ff: int, tp: Optional[TPGroup] = None,
                 dtype=None):
        super().__init__()
        self.up = ColumnParallelLinear(d_model, d_ff, tp=tp, dtype=dtype)
        self.down = RowParallelLinear(d_ff, d_model, tp=tp, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down(torch.nn.functional.gelu(self.up(x), approximate="tanh"))
