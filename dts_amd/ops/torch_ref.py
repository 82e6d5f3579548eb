"""Pure-PyTorch fp32 reference implementations of every custom op.

These define the semantics the HIP/CDNA4 kernels (dts_amd/ops/hip/) must
match bit-for-tolerance; kernel numerics tests compare against these run in
fp32 (tests/ops/). They are also the CPU execution path for the plumbing
config (BASELINE.json config 1) — never the GPU path.
"""

from __future__ import annotations

import math
from typing import Optional

import torch


# ---------------------------------------------------------------------------
# Normalization
# ---------------------------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """y = x / rms(x) * weight, computed in fp32 (Llama RMSNorm)."""
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps)
    return (y * weight.float()).to(dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
):
    """residual' = x + residual; y = rmsnorm(residual'). Returns (y, residual')."""
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


def layernorm(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float = 1e-5
) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    mu = xf.mean(dim=-1, keepdim=True)
    var = (xf - mu).pow(2).mean(dim=-1, keepdim=True)
    y = (xf - mu) * torch.rsqrt(var + eps)
    return (y * weight.float() + bias.float()).to(dtype)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

def build_rope_cache(
    head_dim: int, max_position: int, theta: float, device="cpu", dtype=torch.float32
):
    """cos/sin tables [max_position, head_dim/2] (Llama rotate-half pairing:
    dims (i, i + D/2) rotate together)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, device=device).float() / head_dim)
    )
    t = torch.arange(max_position, device=device).float()
    freqs = torch.outer(t, inv_freq)  # [P, D/2]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def rope_apply(
    q: torch.Tensor,  # [T, Hq, D]
    k: torch.Tensor,  # [T, Hkv, D]
    positions: torch.Tensor,  # [T]
    cos: torch.Tensor,  # [P, D/2]
    sin: torch.Tensor,
):
    """Rotate-half RoPE applied in fp32; returns new (q, k)."""

    def rot(x):
        xf = x.float()
        d2 = x.shape[-1] // 2
        c = cos[positions].unsqueeze(1).float()  # [T, 1, D/2]
        s = sin[positions].unsqueeze(1).float()
        x1, x2 = xf[..., :d2], xf[..., d2:]
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)

    return rot(q), rot(k)


# ---------------------------------------------------------------------------
# Paged KV append
# ---------------------------------------------------------------------------

def kv_append(
    k: torch.Tensor,  # [T, Hkv, D]
    v: torch.Tensor,
    k_cache: torch.Tensor,  # [num_blocks, Hkv, block_size, D]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [T] flat slot = block_id*block_size + offset
) -> None:
    block_size = k_cache.shape[2]
    blocks = torch.div(slot_mapping, block_size, rounding_mode="floor")
    offs = slot_mapping - blocks * block_size
    k_cache[blocks, :, offs] = k.to(k_cache.dtype)
    v_cache[blocks, :, offs] = v.to(v_cache.dtype)


def gather_kv(
    k_cache: torch.Tensor,  # [num_blocks, Hkv, block_size, D]
    v_cache: torch.Tensor,
    block_table: torch.Tensor,  # [n_blocks_for_seq]
    kv_len: int,
):
    """Contiguous [kv_len, Hkv, D] K/V for one sequence (reference only)."""
    bs = k_cache.shape[2]
    n_blocks = (kv_len + bs - 1) // bs
    k = k_cache[block_table[:n_blocks]]  # [n, Hkv, bs, D]
    v = v_cache[block_table[:n_blocks]]
    k = k.permute(0, 2, 1, 3).reshape(n_blocks * bs, k_cache.shape[1], -1)[:kv_len]
    v = v.permute(0, 2, 1, 3).reshape(n_blocks * bs, v_cache.shape[1], -1)[:kv_len]
    return k, v


# ---------------------------------------------------------------------------
# Attention (paged, varlen)
# ---------------------------------------------------------------------------

def _sdpa(q, k, v, scale, causal_offset=None):
    """q [Tq, Hq, D], k/v [Tk, Hkv, D] fp32 attention with GQA broadcast.

    causal_offset: positions [Tq] of queries in the sequence; query i may
    attend keys [0, pos_i].
    """
    Tq, Hq, D = q.shape
    Tk, Hkv, _ = k.shape
    rep = Hq // Hkv
    qf = q.float().permute(1, 0, 2)  # [Hq, Tq, D]
    kf = k.float().permute(1, 0, 2)  # [Hkv, Tk, D]
    vf = v.float().permute(1, 0, 2)
    if rep > 1:
        kf = kf.repeat_interleave(rep, dim=0)
        vf = vf.repeat_interleave(rep, dim=0)
    scores = torch.bmm(qf, kf.transpose(1, 2)) * scale  # [Hq, Tq, Tk]
    if causal_offset is not None:
        key_idx = torch.arange(Tk, device=q.device).view(1, 1, Tk)
        qpos = causal_offset.view(1, Tq, 1)
        scores = scores.masked_fill(key_idx > qpos, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.bmm(probs, vf)  # [Hq, Tq, D]
    return out.permute(1, 0, 2).to(q.dtype)


def attn_prefill_paged(
    q: torch.Tensor,  # [Tq_total, Hq, D] prefill queries, seqs concatenated
    cu_q: torch.Tensor,  # [P+1] cumulative query counts
    q_positions: torch.Tensor,  # [Tq_total] absolute position of each query
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [P, max_blocks]
    kv_lens: torch.Tensor,  # [P] total kv length per seq (incl. new tokens)
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Chunked-prefill attention: each query attends to cached prefix + the
    causal part of its own chunk. KV must already be appended to the cache."""
    D = q.shape[-1]
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    P = int(cu_q.shape[0]) - 1
    for i in range(P):
        s, e = int(cu_q[i]), int(cu_q[i + 1])
        kv_len = int(kv_lens[i])
        k, v = gather_kv(k_cache, v_cache, block_tables[i], kv_len)
        out[s:e] = _sdpa(q[s:e], k, v, scale, causal_offset=q_positions[s:e])
    return out


def attn_decode_paged(
    q: torch.Tensor,  # [B, Hq, D] one query token per sequence
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_blocks]
    kv_lens: torch.Tensor,  # [B]
    scale: Optional[float] = None,
) -> torch.Tensor:
    D = q.shape[-1]
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for i in range(q.shape[0]):
        kv_len = int(kv_lens[i])
        k, v = gather_kv(k_cache, v_cache, block_tables[i], kv_len)
        out[i : i + 1] = _sdpa(q[i : i + 1], k, v, scale)
    return out


# ---------------------------------------------------------------------------
# Activations
# ---------------------------------------------------------------------------

def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    """Input [T, 2*I] = [gate | up]; returns silu(gate) * up (SwiGLU)."""
    gate, up = gate_up.chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate_up.dtype)


def gelu(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x.float(), approximate="tanh").to(x.dtype)


# ---------------------------------------------------------------------------
# Sampling
# ---------------------------------------------------------------------------

_MASK64 = (1 << 64) - 1


def mix_seed(base: int, pos: int) -> int:
    """Stateless splitmix64-style seed mix over (base seed, token index).

    Bit-identical to the HIP device function mix_seed64
    (ops/hip/sampling.hip): the host per-step path and the on-device
    chained-decode path must draw the same token for the same position.
    """
    x = (base ^ (pos * 0x9E3779B97F4A7C15)) & _MASK64
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & _MASK64
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & _MASK64
    x ^= x >> 31
    return x & 0x7FFFFFFF


def derive_seeds(
    bases: torch.Tensor, positions: torch.Tensor
) -> torch.Tensor:
    """CPU reference of the derive_seeds kernel: mix(base, pos + 1)."""
    return torch.tensor(
        [
            mix_seed(int(b), int(p) + 1)
            for b, p in zip(bases.tolist(), positions.tolist())
        ],
        dtype=torch.long,
    )


def moe_grouped_linear(
    x: torch.Tensor,  # [P, K]
    w: torch.Tensor,  # [E, N, K]
    counts: torch.Tensor,  # [E]
    offsets: torch.Tensor,  # [E]
) -> torch.Tensor:
    """Reference of the grouped per-expert GEMM (capture-safe MoE)."""
    P, K = x.shape
    E, N, _ = w.shape
    out = torch.zeros(P, N, dtype=x.dtype, device=x.device)
    for e in range(E):
        c = int(counts[e])
        if c == 0:
            continue
        o = int(offsets[e])
        out[o : o + c] = (x[o : o + c].float() @ w[e].float().T).to(x.dtype)
    return out


def top_p_sample(
    logits: torch.Tensor,  # [B, V] fp32
    temperatures: torch.Tensor,  # [B]
    top_ps: torch.Tensor,  # [B]
    generators: Optional[list] = None,
) -> torch.Tensor:
    """Temperature + nucleus sampling; temperature<=0 means greedy."""
    B, V = logits.shape
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    for i in range(B):
        t = float(temperatures[i])
        if t <= 0.0:
            out[i] = int(torch.argmax(logits[i]))
            continue
        probs = torch.softmax(logits[i].float() / t, dim=-1)
        p = float(top_ps[i])
        if p < 1.0:
            sorted_probs, sorted_idx = torch.sort(probs, descending=True)
            cum = torch.cumsum(sorted_probs, dim=-1)
            # keep tokens while cumulative (exclusive) < p
            mask = (cum - sorted_probs) < p
            mask[0] = True
            kept = sorted_probs * mask
            kept = kept / kept.sum()
            gen = generators[i] if generators else None
            pick = torch.multinomial(kept, 1, generator=gen)
            out[i] = sorted_idx[pick]
        else:
            gen = generators[i] if generators else None
            out[i] = torch.multinomial(probs, 1, generator=gen)
    return out
