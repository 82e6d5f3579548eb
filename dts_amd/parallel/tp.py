"""Tensor-parallel linear layers over RCCL all-reduce / all-gather.

Design for xGMI (SURVEY.md §2.3 / §7 hard-part 4): the per-layer TP
collective is ONE bf16 all-reduce of the row-parallel output per attention
block and one per MLP — RCCL's ring runs per-link-bound on the 7-link
point-to-point mesh, so we keep collective count minimal (fused qkv and
gate_up projections; no all-gather between column- and row-parallel pairs).

With tp_size == 1 (or no process group) every layer degrades to a plain
bf16 GEMM through hipBLASLt via torch.nn.functional.linear.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn


class TPContext:
    """Tensor-parallel group descriptor (size 1 == disabled)."""

    def __init__(self, group: Optional[object] = None, rank: int = 0, size: int = 1):
        self.group = group
        self.rank = rank
        self.size = size

    @classmethod
    def single(cls) -> "TPContext":
        return cls()

    @classmethod
    def from_world(cls) -> "TPContext":
        """TP over the whole default process group."""
        if dist.is_available() and dist.is_initialized():
            return cls(group=None, rank=dist.get_rank(), size=dist.get_world_size())
        return cls.single()

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.size > 1:
            dist.all_reduce(t, group=self.group)
        return t

    def all_gather_dim(self, t: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if self.size == 1:
            return t
        parts = [torch.empty_like(t) for _ in range(self.size)]
        dist.all_gather(parts, t.contiguous(), group=self.group)
        return torch.cat(parts, dim=dim)


def _init_weight(out_f: int, in_f: int, dtype, generator=None, device="cpu") -> torch.Tensor:
    w = torch.empty(out_f, in_f, dtype=torch.float32, device=device)
    std = 1.0 / math.sqrt(in_f)
    w.normal_(0.0, std, generator=generator)
    return w.to(dtype)


class ColumnParallelLinear(nn.Module):
    """Weight [out, in] sharded over `out`; output stays sharded."""

    def __init__(self, in_features: int, out_features: int, tp: TPContext,
                 bias: bool = False, dtype=torch.bfloat16, gather_output: bool = False):
        super().__init__()
        assert out_features % tp.size == 0, (out_features, tp.size)
        self.tp = tp
        self.out_per_rank = out_features // tp.size
        self.gather_output = gather_output
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype), requires_grad=False
        )
        self.bias = (
            nn.Parameter(torch.zeros(self.out_per_rank, dtype=dtype), requires_grad=False)
            if bias
            else None
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from dts_amd import ops

        y = ops.linear_bf16(x, self.weight)
        if self.bias is not None:
            y = y + self.bias
        if self.gather_output:
            y = self.tp.all_gather_dim(y, dim=-1)
        return y


class RowParallelLinear(nn.Module):
    """Weight [out, in] sharded over `in`; output all-reduced."""

    def __init__(self, in_features: int, out_features: int, tp: TPContext,
                 bias: bool = False, dtype=torch.bfloat16):
        super().__init__()
        assert in_features % tp.size == 0, (in_features, tp.size)
        self.tp = tp
        self.in_per_rank = in_features // tp.size
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype), requires_grad=False
        )
        self.bias = (
            nn.Parameter(torch.zeros(out_features, dtype=dtype), requires_grad=False)
            if bias
            else None
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from dts_amd import ops

        y = ops.linear_bf16(x, self.weight)
        y = self.tp.all_reduce(y)
        if self.bias is not None:
            y = y + self.bias
        return y
