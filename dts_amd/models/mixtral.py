"""Mixtral-family MoE model (judge config — BASELINE.json config 5).

Attention path is identical to Llama (shared code path, paged KV); the MLP
is a top-k routed mixture of SwiGLU experts. Expert compute is a grouped
GEMM: tokens are sorted by expert and each expert's slice runs one bf16
GEMM (hipBLASLt); EP over RCCL all-to-all is the round-2 extension — with
tp.size>1 experts are sharded over ranks and the combine is the existing
row-parallel all-reduce.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from dts_amd import ops
from dts_amd.models.config import ModelSpec
from dts_amd.models.llama import LlamaAttention
from dts_amd.ops.torch_ref import build_rope_cache
from dts_amd.parallel.tp import TPContext, _init_weight
from dts_amd.serving.batch import ForwardBatch


class MoEMLP(nn.Module):
    """Top-k routed SwiGLU experts, expert-sharded over the TP group."""

    def __init__(self, spec: ModelSpec, tp: TPContext, dtype):
        super().__init__()
        assert spec.num_experts % tp.size == 0
        self.tp = tp
        self.num_experts = spec.num_experts
        self.experts_local = spec.num_experts // tp.size
        self.expert_offset = tp.rank * self.experts_local
        self.top_k = spec.experts_per_token
        self.hidden = spec.hidden_size
        self.inter = spec.intermediate_size
        self.router_w = nn.Parameter(
            torch.empty(spec.num_experts, spec.hidden_size, dtype=dtype),
            requires_grad=False,
        )
        # local experts: [E_local, 2I, H] and [E_local, H, I]
        self.gate_up_w = nn.Parameter(
            torch.empty(self.experts_local, 2 * self.inter, self.hidden, dtype=dtype),
            requires_grad=False,
        )
        self.down_w = nn.Parameter(
            torch.empty(self.experts_local, self.hidden, self.inter, dtype=dtype),
            requires_grad=False,
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T = x.shape[0]
        logits = F.linear(x.float(), self.router_w.float())  # [T, E]
        weights, experts = torch.topk(torch.softmax(logits, dim=-1), self.top_k)
        weights = weights / weights.sum(dim=-1, keepdim=True)

        flat_expert = experts.reshape(-1)  # [T*k]
        flat_tok = (
            torch.arange(T, device=x.device).unsqueeze(1).expand(T, self.top_k).reshape(-1)
        )
        flat_w = weights.reshape(-1)
        if self.tp.size > 1:
            return self._forward_ep(x, flat_tok, flat_expert, flat_w)

        if x.is_cuda and T * self.top_k <= 64 and ops.hip_available():
            # capture-safe decode path: sort assignments by expert on
            # device and run the grouped skinny GEMM with device-side
            # counts — no nonzero()/masked gather, so the whole step is
            # hipGraph-capturable (ADVICE round-1 high)
            return self._forward_grouped(
                x, flat_tok, flat_expert, weights, T
            )

        out = torch.zeros(T, self.hidden, dtype=torch.float32, device=x.device)
        # prefill path: one GEMM per locally-resident expert slice.
        # No host-side `sel.any()` early-outs: a bool() on a device tensor
        # is a stream sync per expert per layer; zero-row GEMMs are free.
        for e_local in range(self.experts_local):
            e = self.expert_offset + e_local
            sel = flat_expert == e
            toks = flat_tok[sel]
            xe = x[toks]
            gu = F.linear(xe, self.gate_up_w[e_local])
            ye = F.linear(ops.silu_mul(gu), self.down_w[e_local])
            out.index_add_(0, toks, ye.float() * flat_w[sel].unsqueeze(1).float())
        return out.to(x.dtype)

    def _forward_grouped(self, x, flat_tok, flat_expert, weights, T):
        """Decode MoE via the grouped per-expert MFMA GEMM: every shape
        is static in T*k, all routing state stays on device."""
        P = flat_tok.shape[0]
        order = torch.argsort(flat_expert, stable=True)
        inv = torch.empty_like(order)
        inv.scatter_(
            0, order, torch.arange(P, device=x.device, dtype=order.dtype)
        )
        # capture-safe histogram: torch.bincount syncs the stream
        # internally (aborted hipGraph capture mid-bench; caught by
        # test_grouped_path_is_capture_safe) — index_add_ does not
        counts = torch.zeros(
            self.num_experts, dtype=torch.int32, device=x.device
        )
        counts.index_add_(
            0, flat_expert, torch.ones_like(flat_expert, dtype=torch.int32)
        )
        offsets = (torch.cumsum(counts, 0) - counts).to(torch.int32)
        x_sorted = x.index_select(0, flat_tok.index_select(0, order))
        gu = ops.moe_grouped_linear(x_sorted, self.gate_up_w, counts, offsets)
        y = ops.moe_grouped_linear(
            ops.silu_mul(gu), self.down_w, counts, offsets
        )
        y = y.index_select(0, inv).view(T, self.top_k, self.hidden)
        out = (y.float() * weights.unsqueeze(-1).float()).sum(dim=1)
        return out.to(x.dtype)

    def _forward_ep(self, x, flat_tok, flat_expert, flat_w) -> torch.Tensor:
        """Expert-parallel dispatch: ship each (token, expert) assignment
        to the owning rank over all-to-all, compute there, ship back
        (SURVEY.md §2.3 'RCCL all-to-all (MoE expert dispatch)')."""
        from dts_amd.parallel.ep import all_to_all_rows, exchange_splits

        T = x.shape[0]
        world = self.tp.size
        dest = torch.div(flat_expert, self.experts_local, rounding_mode="floor")
        order = torch.argsort(dest, stable=True)
        send_splits = [int((dest == r).sum()) for r in range(world)]
        sorted_tok = flat_tok[order]
        sorted_expert = flat_expert[order]
        send_x = x[sorted_tok]

        recv_splits = exchange_splits(send_splits, self.tp.group)
        recv_x = all_to_all_rows(send_x, send_splits, recv_splits, self.tp.group)
        # ship the expert ids alongside (same splits, 1 column)
        recv_e = all_to_all_rows(
            sorted_expert.unsqueeze(1).to(x.dtype),
            send_splits,
            recv_splits,
            self.tp.group,
        ).squeeze(1).long()

        ye = torch.zeros_like(recv_x, dtype=torch.float32)
        for e_local in range(self.experts_local):
            e = self.expert_offset + e_local
            sel = recv_e == e
            gu = F.linear(recv_x[sel], self.gate_up_w[e_local])
            ye[sel] = F.linear(ops.silu_mul(gu), self.down_w[e_local]).float()

        back = all_to_all_rows(
            ye.to(x.dtype), recv_splits, send_splits, self.tp.group
        )
        out = torch.zeros(T, self.hidden, dtype=torch.float32, device=x.device)
        out.index_add_(
            0, sorted_tok, back.float() * flat_w[order].unsqueeze(1).float()
        )
        return out.to(x.dtype)


class MixtralLayer(nn.Module):
    def __init__(self, spec: ModelSpec, tp: TPContext, dtype):
        super().__init__()
        self.input_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.post_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.attn = LlamaAttention(spec, tp, dtype)
        self.moe = MoEMLP(spec, tp, dtype)
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
        hidden = self.moe(hidden)
        return hidden, residual


class MixtralModel(nn.Module):
    arch = "mixtral"

    @property
    def graph_capturable(self) -> bool:
        """Single-rank decode routes through the grouped per-expert GEMM
        (all routing state on device — capture-safe); the EP all-to-all
        path still computes host-side splits and must stay eager."""
        from dts_amd import ops as _ops

        return self.tp.size == 1 and _ops.hip_available()

    def __init__(
        self,
        spec: ModelSpec,
        tp: Optional[TPContext] = None,
        dtype: torch.dtype = torch.bfloat16,
        device: str = "cpu",
    ):
        super().__init__()
        tp = tp or TPContext.single()
        # attention TP shards heads; MoE shards experts over the same group
        self.spec = spec
        self.tp = tp
        self.dtype = dtype
        self.num_kv_heads_local = max(1, spec.num_kv_heads // tp.size)
        self.embed = nn.Parameter(
            torch.empty(spec.vocab_size, spec.hidden_size, dtype=dtype),
            requires_grad=False,
        )
        self.layers = nn.ModuleList(
            [MixtralLayer(spec, tp, dtype) for _ in range(spec.num_layers)]
        )
        self.final_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        from dts_amd.parallel.tp import ColumnParallelLinear

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
        g = torch.Generator().manual_seed(seed)
        spec = self.spec
        with torch.no_grad():
            self.embed.copy_(
                torch.randn(spec.vocab_size, spec.hidden_size, generator=g)
                .mul_(0.02)
                .to(self.dtype)
            )
            for layer in self.layers:
                for lin in (layer.attn.qkv_proj, layer.attn.o_proj):
                    lin.weight.copy_(
                        _init_weight(*lin.weight.shape, dtype=self.dtype, generator=g)
                    )
                layer.moe.router_w.copy_(
                    _init_weight(*layer.moe.router_w.shape, dtype=self.dtype, generator=g)
                )
                for w in (layer.moe.gate_up_w, layer.moe.down_w):
                    for e in range(w.shape[0]):
                        w[e].copy_(
                            _init_weight(*w[e].shape, dtype=self.dtype, generator=g)
                        )
            self.lm_head.weight.copy_(
                _init_weight(*self.lm_head.weight.shape, dtype=self.dtype, generator=g)
            )

    def forward(self, batch: ForwardBatch, kv_pool) -> torch.Tensor:
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
