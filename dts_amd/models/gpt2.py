"""GPT-2 model (plumbing config — BASELINE.json config 1, CPU).

Same flat-batch + paged-KV execution as Llama (models/llama.py); the
architectural deltas are learned absolute position embeddings, pre-LN
LayerNorm with biases, fused-QKV with bias, GELU MLP and tied embeddings.
No TP (this model exists to prove the engine loop end-to-end on CPU).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from dts_amd import ops
from dts_amd.models.config import ModelSpec
from dts_amd.serving.batch import ForwardBatch


class GPT2Layer(nn.Module):
    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        h = spec.hidden_size
        self.ln1_w = nn.Parameter(torch.ones(h, dtype=dtype), requires_grad=False)
        self.ln1_b = nn.Parameter(torch.zeros(h, dtype=dtype), requires_grad=False)
        self.ln2_w = nn.Parameter(torch.ones(h, dtype=dtype), requires_grad=False)
        self.ln2_b = nn.Parameter(torch.zeros(h, dtype=dtype), requires_grad=False)
        self.qkv_w = nn.Parameter(torch.empty(3 * h, h, dtype=dtype), requires_grad=False)
        self.qkv_b = nn.Parameter(torch.zeros(3 * h, dtype=dtype), requires_grad=False)
        self.o_w = nn.Parameter(torch.empty(h, h, dtype=dtype), requires_grad=False)
        self.o_b = nn.Parameter(torch.zeros(h, dtype=dtype), requires_grad=False)
        self.fc_w = nn.Parameter(
            torch.empty(spec.intermediate_size, h, dtype=dtype), requires_grad=False
        )
        self.fc_b = nn.Parameter(
            torch.zeros(spec.intermediate_size, dtype=dtype), requires_grad=False
        )
        self.proj_w = nn.Parameter(
            torch.empty(h, spec.intermediate_size, dtype=dtype), requires_grad=False
        )
        self.proj_b = nn.Parameter(torch.zeros(h, dtype=dtype), requires_grad=False)
        self.num_heads = spec.num_heads
        self.head_dim = spec.head_dim
        self.scale = 1.0 / math.sqrt(spec.head_dim)
        self.eps = spec.layernorm_eps

    def forward(self, hidden, batch: ForwardBatch, kv_layer):
        T = hidden.shape[0]
        x = ops.layernorm(hidden, self.ln1_w, self.ln1_b, self.eps)
        qkv = F.linear(x, self.qkv_w, self.qkv_b)
        q, k, v = qkv.chunk(3, dim=-1)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_heads, self.head_dim)
        v = v.view(T, self.num_heads, self.head_dim)
        k_cache, v_cache = kv_layer
        ops.kv_append(k, v, k_cache, v_cache, batch.slot_mapping)

        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        np_tok = batch.num_prefill_tokens
        if batch.num_prefill_seqs:
            ops.attn_prefill_paged(
                q[:np_tok], batch.cu_q, batch.positions[:np_tok],
                k_cache, v_cache, batch.prefill_block_tables,
                batch.prefill_kv_lens, self.scale, out=out[:np_tok],
            )
        if batch.num_decode_seqs:
            ops.attn_decode_paged(
                q[np_tok:], k_cache, v_cache,
                batch.decode_block_tables, batch.decode_kv_lens, self.scale,
                out=out[np_tok:],
            )
        hidden = hidden + F.linear(out.reshape(T, -1), self.o_w, self.o_b)
        x = ops.layernorm(hidden, self.ln2_w, self.ln2_b, self.eps)
        x = ops.gelu(F.linear(x, self.fc_w, self.fc_b))
        return hidden + F.linear(x, self.proj_w, self.proj_b)


class GPT2Model(nn.Module):
    arch = "gpt2"

    def __init__(self, spec: ModelSpec, dtype=torch.float32, device="cpu"):
        super().__init__()
        self.spec = spec
        self.dtype = dtype
        self.wte = nn.Parameter(
            torch.empty(spec.vocab_size, spec.hidden_size, dtype=dtype),
            requires_grad=False,
        )
        self.wpe = nn.Parameter(
            torch.empty(spec.max_position, spec.hidden_size, dtype=dtype),
            requires_grad=False,
        )
        self.layers = nn.ModuleList(
            [GPT2Layer(spec, dtype) for _ in range(spec.num_layers)]
        )
        self.lnf_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.lnf_b = nn.Parameter(
            torch.zeros(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.to(device)

    def random_init(self, seed: int = 0) -> None:
        g = torch.Generator().manual_seed(seed)

        def init_(p, std=0.02):
            with torch.no_grad():
                p.copy_(
                    torch.randn(*p.shape, generator=g).mul_(std).to(p.dtype)
                )

        init_(self.wte)
        init_(self.wpe, 0.01)
        for layer in self.layers:
            for w in (layer.qkv_w, layer.o_w, layer.fc_w, layer.proj_w):
                init_(w, 1.0 / math.sqrt(w.shape[1]))

    def forward(self, batch: ForwardBatch, kv_pool) -> torch.Tensor:
        hidden = self.wte[batch.token_ids] + self.wpe[batch.positions]
        for i, layer in enumerate(self.layers):
            hidden = layer(hidden, batch, kv_pool.layer(i))
        if batch.sample_indices is not None:
            hidden = hidden[batch.sample_indices]
        hidden = ops.layernorm(hidden, self.lnf_w, self.lnf_b, self.spec.layernorm_eps)
        return F.linear(hidden, self.wte).float()  # tied embeddings
