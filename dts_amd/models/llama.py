"""Llama-family model (Llama-3-8B/70B and tiny test variants).

Executes over the flat mixed prefill+decode batch (serving/batch.py) with
a paged KV cache. Hot ops route through dts_amd.ops (HIP kernels on GPU,
torch fp32 references on CPU); plain projections are bf16 GEMMs through
hipBLASLt (torch F.linear). Logits are computed only for the rows being
sampled — on the 6x5 search workload most steps sample O(batch) rows out
of O(thousands) prefill tokens, so this skips most of the
[T,4096]x[4096,128256] lm_head work.

TP: fused qkv / gate_up are column-parallel, o_proj / down_proj
row-parallel (one RCCL all-reduce each per layer — dts_amd/parallel/tp.py);
heads are sharded so attention kernels are TP-oblivious.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from dts_amd import ops
from dts_amd.models.config import ModelSpec
from dts_amd.ops.torch_ref import build_rope_cache
from dts_amd.parallel.tp import (
    ColumnParallelLinear,
    RowParallelLinear,
    TPContext,
    _init_weight,
)
from dts_amd.serving.batch import ForwardBatch


class LlamaAttention(nn.Module):
    def __init__(self, spec: ModelSpec, tp: TPContext, dtype):
        super().__init__()
        self.tp = tp
        assert spec.num_heads % tp.size == 0
        assert spec.num_kv_heads % tp.size == 0 or tp.size % spec.num_kv_heads == 0
        self.num_heads = spec.num_heads // tp.size
        self.num_kv_heads = max(1, spec.num_kv_heads // tp.size)
        self.head_dim = spec.head_dim
        self.scale = 1.0 / math.sqrt(spec.head_dim)
        q_out = self.num_heads * self.head_dim * tp.size
        kv_out = self.num_kv_heads * self.head_dim * tp.size
        self.qkv_proj = ColumnParallelLinear(
            spec.hidden_size, q_out + 2 * kv_out, tp, dtype=dtype
        )
        self.o_proj = RowParallelLinear(
            self.num_heads * self.head_dim * tp.size, spec.hidden_size, tp, dtype=dtype
        )
        self.q_size = self.num_heads * self.head_dim
        self.kv_size = self.num_kv_heads * self.head_dim

    def forward(
        self,
        hidden: torch.Tensor,  # [T, H]
        batch: ForwardBatch,
        kv_layer: tuple,  # (k_cache, v_cache) for this layer
        rope: tuple,  # (cos, sin)
    ) -> torch.Tensor:
        T = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size], dim=-1)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_kv_heads, self.head_dim)
        v = v.view(T, self.num_kv_heads, self.head_dim)

        k_cache, v_cache = kv_layer
        cos, sin = rope
        q, k = ops.rope_kv_append(
            q, k, v, batch.positions, cos, sin, k_cache, v_cache, batch.slot_mapping
        )

        # q may be a strided view of qkv; output is always dense. The
        # kernels write straight into the dense slices (dim-0 slices of a
        # contiguous tensor) — a slice ASSIGNMENT here cost a D2D
        # copyBuffer per section per layer (2.7% of GPU busy, r2 profile)
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        np_tok = batch.num_prefill_tokens
        if batch.num_prefill_seqs:
            ops.attn_prefill_paged(
                q[:np_tok],
                batch.cu_q,
                batch.positions[:np_tok],
                k_cache,
                v_cache,
                batch.prefill_block_tables,
                batch.prefill_kv_lens,
                self.scale,
                out=out[:np_tok],
            )
        if batch.num_decode_seqs:
            ops.attn_decode_paged(
                q[np_tok:],
                k_cache,
                v_cache,
                batch.decode_block_tables,
                batch.decode_kv_lens,
                self.scale,
                out=out[np_tok:],
            )
        return self.o_proj(out.reshape(T, -1))


class LlamaMLP(nn.Module):
    def __init__(self, spec: ModelSpec, tp: TPContext, dtype):
        super().__init__()
        self.gate_up_proj = ColumnParallelLinear(
            spec.hidden_size, 2 * spec.intermediate_size, tp, dtype=dtype
        )
        self.down_proj = RowParallelLinear(
            spec.intermediate_size, spec.hidden_size, tp, dtype=dtype
        )
        self.inter_per_rank = spec.intermediate_size // tp.size

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        gu = self.gate_up_proj(x)
        # column-parallel layout: [gate_shard | up_shard] per rank
        return self.down_proj(ops.silu_mul(gu))


class LlamaLayer(nn.Module):
    def __init__(self, spec: ModelSpec, tp: TPContext, dtype):
        super().__init__()
        self.input_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.post_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.attn = LlamaAttention(spec, tp, dtype)
        self.mlp = LlamaMLP(spec, tp, dtype)
        self.eps = spec.rms_eps

    def forward(self, hidden, residual, batch, kv_layer, rope):
        if residual is None:
            residual = hidden
            hidden = ops.rmsnorm(hidden, self.input_norm_w, self.eps)
        else:
            hidden, residual = ops.fused_add_rmsnorm(
                hidden, residual, self.input_norm_w, self.eps
            )
        hidden = self.attn(hidden, batch, kv_layer, rope)
        hidden, residual = ops.fused_add_rmsnorm(
            hidden, residual, self.post_norm_w, self.eps
        )
        hidden = self.mlp(hidden)
        return hidden, residual


class LlamaModel(nn.Module):
    arch = "llama"

    def __init__(
        self,
        spec: ModelSpec,
        tp: Optional[TPContext] = None,
        dtype: torch.dtype = torch.bfloat16,
        device: str = "cpu",
    ):
        super().__init__()
        tp = tp or TPContext.single()
        self.spec = spec
        self.tp = tp
        self.dtype = dtype
        self.num_kv_heads_local = max(1, spec.num_kv_heads // tp.size)
        self.embed = nn.Parameter(
            torch.empty(spec.vocab_size, spec.hidden_size, dtype=dtype),
            requires_grad=False,
        )
        self.layers = nn.ModuleList(
            [LlamaLayer(spec, tp, dtype) for _ in range(spec.num_layers)]
        )
        self.final_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.lm_head = ColumnParallelLinear(
            spec.hidden_size, spec.vocab_size, tp, dtype=dtype, gather_output=True
        )
        cos, sin = build_rope_cache(
            spec.head_dim, spec.max_position, spec.rope_theta, dtype=torch.float32
        )
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.to(device)

    def random_init(self, seed: int = 0) -> None:
        """Deterministic random init at the right scale (synthetic bench).

        Generates directly on the model's device — initializing 8B params
        on a GPU is near-instant vs ~1 min through host randn.
        """
        dev = self.embed.device
        g = torch.Generator(device=dev).manual_seed(seed)
        spec = self.spec
        with torch.no_grad():
            self.embed.copy_(
                torch.randn(
                    spec.vocab_size, spec.hidden_size, generator=g, device=dev
                )
                .mul_(0.02)
                .to(self.dtype)
            )
            for layer in self.layers:
                for lin in (
                    layer.attn.qkv_proj,
                    layer.attn.o_proj,
                    layer.mlp.gate_up_proj,
                    layer.mlp.down_proj,
                ):
                    lin.weight.copy_(
                        _init_weight(
                            *lin.weight.shape,
                            dtype=self.dtype,
                            generator=g,
                            device=dev,
                        )
                    )
            self.lm_head.weight.copy_(
                _init_weight(
                    *self.lm_head.weight.shape,
                    dtype=self.dtype,
                    generator=g,
                    device=dev,
                )
            )

    def forward(self, batch: ForwardBatch, kv_pool) -> torch.Tensor:
        """Returns logits [len(sample_indices), vocab]."""
        hidden = self.embed[batch.token_ids]
        residual = None
        rope = (self.rope_cos, self.rope_sin)
        for i, layer in enumerate(self.layers):
            hidden, residual = layer(hidden, residual, batch, kv_pool.layer(i), rope)
        hidden, _ = ops.fused_add_rmsnorm(
            hidden, residual, self.final_norm_w, self.spec.rms_eps
        )
        if batch.sample_indices is not None:
            hidden = hidden[batch.sample_indices]
        return self.lm_head(hidden).float()
