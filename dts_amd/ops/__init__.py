"""Op dispatch: HIP/CDNA4 kernels on GPU, torch fp32 references on CPU.

Policy (round-end audit requirement): on a GPU box the hand-written HIP
extension MUST be the path that runs — if a CUDA tensor reaches an op and
the extension is not importable, we raise instead of silently falling back
to eager PyTorch. Set DTS_ALLOW_TORCH_FALLBACK=1 only for bring-up debugging.
"""

from __future__ import annotations

import os

import torch

from dts_amd.ops import torch_ref
from dts_amd.utils.logging import logger

_hip = None
_hip_load_error: Exception | None = None


def _try_load_hip():
    global _hip, _hip_load_error
    if _hip is not None or _hip_load_error is not None:
        return _hip
    try:
        from dts_amd.ops import _hip_ext_loader

        _hip = _hip_ext_loader.load()
        logger.info("HIP extension loaded: %s", _hip.__name__)
    except Exception as e:  # noqa: BLE001
        _hip_load_error = e
    return _hip


def hip_available() -> bool:
    return _try_load_hip() is not None


def _require_hip():
    ext = _try_load_hip()
    if ext is None:
        if os.environ.get("DTS_ALLOW_TORCH_FALLBACK") == "1":
            return None
        raise RuntimeError(
            "dts_amd HIP extension not available on a GPU device "
            f"(load error: {_hip_load_error}); refusing silent eager fallback. "
            "Build with `python -m dts_amd.ops.build` or set "
            "DTS_ALLOW_TORCH_FALLBACK=1 for debugging."
        )
    return ext


def _on_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


# ---------------------------------------------------------------------------
# Public ops — each dispatches on device
# ---------------------------------------------------------------------------

def rmsnorm(x, weight, eps=1e-5):
    if _on_gpu(x):
        ext = _require_hip()
        if ext is not None:
            out = torch.empty_like(x)
            ext.rmsnorm(out, x, weight, eps)
            return out
    return torch_ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps=1e-5):
    if _on_gpu(x):
        ext = _require_hip()
        if ext is not None:
            # in-place: x <- normed, residual <- x + residual
            ext.fused_add_rmsnorm(x, residual, weight, eps)
            return x, residual
    return torch_ref.fused_add_rmsnorm(x, residual, weight, eps)


def layernorm(x, weight, bias, eps=1e-5):
    if _on_gpu(x):
        ext = _require_hip()
        if ext is not None:
            out = torch.empty_like(x)
            ext.layernorm(out, x, weight, bias, eps)
            return out
    return torch_ref.layernorm(x, weight, bias, eps)


def rope_kv_append(q, k, v, positions, cos, sin, k_cache, v_cache, slot_mapping):
    """Fused: RoPE(q,k) in place + append (k,v) into the paged cache.

    Returns (q, k) rotated. One kernel on GPU (saves two round trips over
    HBM for k); reference path composes the two torch ops.
    """
    if _on_gpu(q):
        ext = _require_hip()
        if ext is not None:
            ext.rope_kv_append(q, k, v, positions, cos, sin, k_cache, v_cache, slot_mapping)
            return q, k
    q, k = torch_ref.rope_apply(q, k, positions, cos, sin)
    torch_ref.kv_append(k, v, k_cache, v_cache, slot_mapping)
    return q, k


def kv_append(k, v, k_cache, v_cache, slot_mapping):
    if _on_gpu(k):
        ext = _require_hip()
        if ext is not None:
            ext.kv_append(k, v, k_cache, v_cache, slot_mapping)
            return
    torch_ref.kv_append(k, v, k_cache, v_cache, slot_mapping)


def attn_prefill_paged(
    q, cu_q, q_positions, k_cache, v_cache, block_tables, kv_lens, scale=None,
    out=None,
):
    if _on_gpu(q):
        ext = _require_hip()
        if ext is not None:
            import math

            if out is None or not out.is_contiguous():
                out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
            ext.attn_prefill_paged(
                out, q, cu_q, q_positions, k_cache, v_cache, block_tables, kv_lens,
                scale if scale is not None else 1.0 / math.sqrt(q.shape[-1]),
            )
            return out
    ref = torch_ref.attn_prefill_paged(
        q, cu_q, q_positions, k_cache, v_cache, block_tables, kv_lens, scale
    )
    if out is not None:
        out.copy_(ref)
        return out
    return ref


def attn_decode_paged(q, k_cache, v_cache, block_tables, kv_lens, scale=None, out=None):
    if _on_gpu(q):
        ext = _require_hip()
        if ext is not None:
            import math

            if out is None or not out.is_contiguous():
                out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
            ext.attn_decode_paged(
                out, q, k_cache, v_cache, block_tables, kv_lens,
                scale if scale is not None else 1.0 / math.sqrt(q.shape[-1]),
            )
            return out
    ref = torch_ref.attn_decode_paged(q, k_cache, v_cache, block_tables, kv_lens, scale)
    if out is not None:
        out.copy_(ref)
        return out
    return ref


def silu_mul(gate_up):
    if gate_up.shape[0] == 0:  # zero-row expert slices (MoE grouped GEMMs)
        return gate_up[:, : gate_up.shape[1] // 2]
    if _on_gpu(gate_up):
        ext = _require_hip()
        if ext is not None:
            T = gate_up.shape[0]
            out = torch.empty(
                (T, gate_up.shape[1] // 2), dtype=gate_up.dtype, device=gate_up.device
            )
            ext.silu_mul(out, gate_up)
            return out
    return torch_ref.silu_mul(gate_up)


def gelu(x):
    return torch_ref.gelu(x)


_sampler_ws: dict = {}


def _sampler_workspace(S: int, device):
    """Cached v2 sampler workspace (stable pointers for graph capture)."""
    key = (str(device), S)
    ws = _sampler_ws.get(key)
    if ws is None:
        ws = (
            torch.zeros(S, dtype=torch.int64, device=device),
            torch.zeros(S, 1024, dtype=torch.float32, device=device),
            torch.zeros(S, 16, dtype=torch.float32, device=device),
            torch.zeros(S, dtype=torch.float32, device=device),
        )
        _sampler_ws[key] = ws
    return ws


def top_p_sample(logits, temperatures, top_ps, generators=None, seeds=None, out=None):
    if _on_gpu(logits):
        ext = _require_hip()
        if ext is not None and seeds is not None:
            S, V = logits.shape
            if out is None:
                out = torch.empty(S, dtype=torch.long, device=logits.device)
            # v2 (gridded) at decode widths: the one-WG-per-row v1 left
            # ~250 of 256 CUs idle (364 us avg on the r2 bench); v1 keeps
            # tiny vocabs (guided sub-vocab picks) and very wide batches
            if V >= 4096 and S <= 32 and os.environ.get("DTS_SAMPLER_V1") != "1":
                ws = _sampler_workspace(S, logits.device)
                ext.top_p_sample_v2(
                    out, logits, temperatures, top_ps, seeds, *ws
                )
            else:
                ext.top_p_sample(out, logits, temperatures, top_ps, seeds)
            return out
    return torch_ref.top_p_sample(logits, temperatures, top_ps, generators)


def derive_seeds(out, bases, positions):
    """out[i] = mix_seed(bases[i], positions[i] + 1) — on-device for the
    chained decode loop (no host round-trip per step)."""
    if _on_gpu(out):
        ext = _require_hip()
        if ext is not None:
            ext.derive_seeds(out, bases, positions)
            return out
    out.copy_(torch_ref.derive_seeds(bases, positions))
    return out


mix_seed = torch_ref.mix_seed


def moe_grouped_linear(x, w, counts, offsets):
    """Grouped per-expert GEMM over contiguous row segments (sorted by
    expert): out[p] = x[p] @ w[expert_of(p)]^T. counts/offsets are DEVICE
    tensors, so a MoE decode step stays hipGraph-capturable."""
    if _on_gpu(x):
        ext = _require_hip()
        if ext is not None:
            out = torch.empty(
                (x.shape[0], w.shape[1]), dtype=x.dtype, device=x.device
            )
            ext.moe_grouped_linear(out, x, w, counts, offsets)
            return out
    return torch_ref.moe_grouped_linear(x, w, counts, offsets)


def linear_bf16(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """y = x @ W^T. Skinny decode batches (M<=16, bf16, K%512==0) stream
    through custom weight-streaming kernels; everything else is a
    hipBLASLt GEMM via F.linear.

    Dispatch (measured A/B, profiles/microbench r2 + the r2 bench
    kernel table): hipBLASLt has a ~20 us per-GEMM floor (caps <=96 MB
    weights at ~2.5 TB/s), and INSIDE captured hipGraphs it picks
    capture-unfriendly algorithms — the bench's graph-replayed gate_up
    averaged 59.5 us (~2-4 TB/s) where isolation measured 20.7. The
    skinny MFMA GEMM has no algo selection and measures 22-30 us on the
    117 MB shapes at M=4..16, so it takes EVERY shape at M<=16; the
    VALU GEMV keeps M<=2 on shallow-K big weights (19.7 us vs skinny's
    21.2 on gate_up; it underfills the grid at K=14336)."""
    M = x.shape[0] if x.dim() == 2 else 0
    if (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and x.dim() == 2
        and x.shape[1] % 512 == 0
        and weight.stride(1) == 1
        and 1 <= M <= 16
    ):
        ext = _require_hip()
        if ext is not None:
            out = torch.empty(
                (M, weight.shape[0]), dtype=x.dtype, device=x.device
            )
            big = weight.shape[0] * weight.shape[1] * 2 > (96 << 20)
            if M <= 2 and big and x.shape[1] <= 8192:
                ext.gemv_bf16(out, x, weight)
            else:
                ext.gemm_skinny_bf16(out, x, weight)
            return out
    return torch.nn.functional.linear(x, weight)
