"""Isolated prefill-attention kernel probe (TF + for PMC runs).

One sequence, q_len=N self-attention (causal), Llama-3-8B shapes.
FLOPs = 2 ops x (QK^T + PV) x Hq x N^2/2 x D.
"""

import json
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from dts_amd.ops import _hip_ext_loader  # noqa: E402

ext = _hip_ext_loader.load()


def probe(N=4096, Hq=32, Hkv=8, D=128, iters=20):
    torch.manual_seed(0)
    BS = 16
    n_blocks = N // BS + 2
    q = torch.randn(N, Hq, D, dtype=torch.bfloat16, device="cuda")
    kc = torch.randn(n_blocks, Hkv, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    bt = torch.arange(1, n_blocks + 1, dtype=torch.int32, device="cuda").unsqueeze(0)
    kvl = torch.tensor([N], dtype=torch.int32, device="cuda")
    cu_q = torch.tensor([0, N], dtype=torch.int32, device="cuda")
    q_pos = torch.arange(N, dtype=torch.long, device="cuda")
    out = torch.empty_like(q)
    scale = D ** -0.5
    flops = 2 * 2 * Hq * (N * N / 2) * D
    results = {}
    # numerics FIRST (and flushed) so a kernel fault localizes to its
    # variant instead of discarding block-buffered timing output
    out_ref = torch.empty_like(q)
    ext.attn_prefill_paged(out_ref, q, cu_q, q_pos, kc, vc, bt, kvl, scale, 0)
    torch.cuda.synchronize()
    print(f"numerics N={N}: v1 ok", flush=True)
    for var in (4, 5):
        out_v = torch.empty_like(q)
        ext.attn_prefill_paged(out_v, q, cu_q, q_pos, kc, vc, bt, kvl, scale, var)
        torch.cuda.synchronize()
        err = float((out_ref.float() - out_v.float()).abs().max())
        print(f"numerics N={N}: variant {var} max_abs_err={err}", flush=True)
    # within-probe interleaved A/B (guide §5.4 rule 24): 6 rounds each
    for swz in (0, 4, 5, 0, 4, 5):
        for _ in range(2):
            ext.attn_prefill_paged(out, q, cu_q, q_pos, kc, vc, bt, kvl,
                                   scale, swz)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            ext.attn_prefill_paged(out, q, cu_q, q_pos, kc, vc, bt, kvl,
                                   scale, swz)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        results.setdefault(swz, []).append(round(flops / dt / 1e12, 1))
    # numerics: v2 must match v1 closely (different tile size changes
    # the online-softmax accumulation order, so not bitwise)
    out0 = torch.empty_like(q)
    out1 = torch.empty_like(q)
    ext.attn_prefill_paged(out0, q, cu_q, q_pos, kc, vc, bt, kvl, scale, 0)
    ext.attn_prefill_paged(out1, q, cu_q, q_pos, kc, vc, bt, kvl, scale, 4)
    max_err = float((out0.float() - out1.float()).abs().max())
    print(  # noqa
        json.dumps(
            {
                "probe": f"prefill_attn_N{N}",
                "TF_v1": results[0],
                "TF_v5k32": results.get(4),
                "TF_v5k64": results.get(5),
                "v1_v5_max_abs_err": max_err,
            }
        ),
        flush=True,
    )


if __name__ == "__main__":
    for n in (2048, 4096, 8192):
        probe(N=n)
