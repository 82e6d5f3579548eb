"""HIP kernel numerics vs the plain-PyTorch fp32 references (torch_ref).

Every test builds random bf16 inputs at Llama-3-8B shapes, runs the CDNA4
kernel on the GPU, and compares against dts_amd/ops/torch_ref.py computed
in fp32 from the SAME bf16 values. Tolerances sized for bf16 I/O with
fp32 accumulation. All tests @gpu (run via gpurun on an MI355X).
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from dts_amd.ops import _hip_ext_loader

    ext = _hip_ext_loader.load()
else:  # collected but skipped off-GPU
    ext = None

from dts_amd.ops import torch_ref

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(1234)


class TestRMSNorm:
    @pytest.mark.parametrize("T,H", [(1, 4096), (17, 4096), (333, 4096), (64, 8192)])
    def test_rmsnorm(self, T, H):
        x = bf(torch.randn(T, H)).to(DEV)
        w = bf(torch.randn(H).abs() + 0.5).to(DEV)
        out = torch.empty_like(x)
        ext.rmsnorm(out, x, w, 1e-5)
        ref = torch_ref.rmsnorm(x.cpu().float(), w.cpu().float(), 1e-5)
        torch.testing.assert_close(out.cpu().float(), ref, atol=2e-2, rtol=2e-2)

    def test_fused_add_rmsnorm(self):
        T, H = 129, 4096
        x = bf(torch.randn(T, H)).to(DEV)
        r = bf(torch.randn(T, H)).to(DEV)
        w = bf(torch.randn(H).abs() + 0.5).to(DEV)
        x_ref, r_ref = x.cpu().float(), r.cpu().float()
        ext.fused_add_rmsnorm(x, r, w, 1e-5)
        ref_norm, ref_res = torch_ref.fused_add_rmsnorm(
            x_ref, r_ref, w.cpu().float(), 1e-5
        )
        torch.testing.assert_close(r.cpu().float(), ref_res, atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(x.cpu().float(), ref_norm, atol=3e-2, rtol=3e-2)


class TestLayerNorm:
    def test_layernorm(self):
        T, H = 65, 768
        x = bf(torch.randn(T, H)).to(DEV)
        w = bf(torch.randn(H).abs() + 0.5).to(DEV)
        b = bf(torch.randn(H) * 0.1).to(DEV)
        out = torch.empty_like(x)
        ext.layernorm(out, x, w, b, 1e-5)
        ref = torch_ref.layernorm(
            x.cpu().float(), w.cpu().float(), b.cpu().float(), 1e-5
        )
        torch.testing.assert_close(out.cpu().float(), ref, atol=3e-2, rtol=3e-2)


class TestSiluMul:
    def test_silu_mul(self):
        T, I = 77, 14336
        gu = bf(torch.randn(T, 2 * I)).to(DEV)
        out = torch.empty(T, I, dtype=torch.bfloat16, device=DEV)
        ext.silu_mul(out, gu)
        ref = torch_ref.silu_mul(gu.cpu().float())
        torch.testing.assert_close(out.cpu().float(), ref, atol=2e-2, rtol=2e-2)


class TestRopeKV:
    def test_rope_kv_append_strided_views(self):
        """q/k/v as SPLIT VIEWS of a fused qkv tensor (the model's actual
        layout — token-row stride != H*D). Regression test for the
        stride-blind addressing bug that passed contiguous unit tests but
        corrupted the model path."""
        T, Hq, Hk, D, BS, NB = 33, 32, 8, 128, 16, 8
        qkv = bf(torch.randn(T, (Hq + 2 * Hk) * D)).to(DEV)
        q = qkv[:, : Hq * D].view(T, Hq, D)
        k = qkv[:, Hq * D : (Hq + Hk) * D].view(T, Hk, D)
        v = qkv[:, (Hq + Hk) * D :].view(T, Hk, D)
        assert not q.is_contiguous()
        positions = torch.randint(0, 500, (T,), dtype=torch.long).to(DEV)
        cos, sin = torch_ref.build_rope_cache(D, 1024, 500000.0)
        slots = torch.randperm(NB * BS)[:T].to(torch.long).to(DEV)
        kc = torch.zeros(NB, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
        vc = torch.zeros_like(kc)

        q_ref, k_ref = torch_ref.rope_apply(
            q.cpu().float(), k.cpu().float(), positions.cpu(), cos, sin
        )
        kc_ref = torch.zeros(NB, Hk, BS, D)
        vc_ref = torch.zeros(NB, Hk, BS, D)
        torch_ref.kv_append(k_ref, v.cpu().float(), kc_ref, vc_ref, slots.cpu())

        ext.rope_kv_append(q, k, v, positions, cos.to(DEV), sin.to(DEV), kc, vc, slots)
        torch.testing.assert_close(q.cpu().float(), q_ref, atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(kc.cpu().float(), kc_ref, atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(vc.cpu().float(), vc_ref, atol=2e-2, rtol=2e-2)

        # strided-q attention over the appended cache
        kvl = torch.tensor([T], dtype=torch.int32, device=DEV)
        # build a block table covering the scattered slots? use dense cache
        # instead: run decode attn with q row 0 against a dense table
        # (covered fully by TestDecodeAttention) — here we only check the
        # strided-q prefill read path with a simple dense cache:
        kc2, vc2, bt = make_paged_kv(1, Hk, D, [T])
        cu_q = torch.tensor([0, T], dtype=torch.int32, device=DEV)
        q_pos = torch.arange(T, dtype=torch.long, device=DEV)
        out = torch.empty(T, Hq, D, dtype=torch.bfloat16, device=DEV)
        scale = 1.0 / math.sqrt(D)
        ext.attn_prefill_paged(out, q, cu_q, q_pos, kc2, vc2, bt, kvl, scale)
        ref = torch_ref.attn_prefill_paged(
            q.cpu().float(), cu_q.cpu(), q_pos.cpu(), kc2.cpu().float(),
            vc2.cpu().float(), bt.cpu(), kvl.cpu(), scale,
        )
        torch.testing.assert_close(out.cpu().float(), ref, atol=2.5e-2, rtol=2.5e-2)

    def test_rope_kv_append(self):
        T, Hq, Hk, D, BS, NB = 50, 32, 8, 128, 16, 8
        q = bf(torch.randn(T, Hq, D)).to(DEV)
        k = bf(torch.randn(T, Hk, D)).to(DEV)
        v = bf(torch.randn(T, Hk, D)).to(DEV)
        positions = torch.randint(0, 500, (T,), dtype=torch.long).to(DEV)
        cos, sin = torch_ref.build_rope_cache(D, 1024, 500000.0)
        slots = torch.randperm(NB * BS)[:T].to(torch.long).to(DEV)
        kc = torch.zeros(NB, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
        vc = torch.zeros_like(kc)

        q_ref, k_ref = torch_ref.rope_apply(
            q.cpu().float(), k.cpu().float(), positions.cpu(), cos, sin
        )
        kc_ref = torch.zeros(NB, Hk, BS, D)
        vc_ref = torch.zeros(NB, Hk, BS, D)
        torch_ref.kv_append(k_ref, v.cpu().float(), kc_ref, vc_ref, slots.cpu())

        ext.rope_kv_append(
            q, k, v, positions, cos.to(DEV), sin.to(DEV), kc, vc, slots
        )
        torch.testing.assert_close(q.cpu().float(), q_ref, atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(k.cpu().float(), k_ref, atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(kc.cpu().float(), kc_ref, atol=2e-2, rtol=2e-2)
        torch.testing.assert_close(vc.cpu().float(), vc_ref, atol=2e-2, rtol=2e-2)


def make_paged_kv(B, Hkv, D, kv_lens, BS=16):
    max_blocks = (max(kv_lens) + BS - 1) // BS
    NB = B * max_blocks + 1
    kc = bf(torch.randn(NB, Hkv, BS, D)).to(DEV)
    vc = bf(torch.randn(NB, Hkv, BS, D)).to(DEV)
    bt = torch.zeros(B, max_blocks, dtype=torch.int32)
    used = 1
    for i in range(B):
        nb = (kv_lens[i] + BS - 1) // BS
        bt[i, :nb] = torch.arange(used, used + nb, dtype=torch.int32)
        used += nb
    return kc, vc, bt.to(DEV)


class TestDecodeAttention:
    @pytest.mark.parametrize("Hq,Hkv,D", [(32, 8, 128), (8, 1, 128), (8, 8, 128), (12, 12, 64)])
    def test_decode(self, Hq, Hkv, D):
        B = 9
        kv_lens = [1, 5, 16, 17, 63, 64, 100, 255, 1000][:B]
        q = bf(torch.randn(B, Hq, D)).to(DEV)
        kc, vc, bt = make_paged_kv(B, Hkv, D, kv_lens)
        kvl = torch.tensor(kv_lens, dtype=torch.int32, device=DEV)
        out = torch.empty_like(q)
        scale = 1.0 / math.sqrt(D)
        ext.attn_decode_paged(out, q, kc, vc, bt, kvl, scale)
        ref = torch_ref.attn_decode_paged(
            q.cpu().float(), kc.cpu().float(), vc.cpu().float(), bt.cpu(),
            kvl.cpu(), scale,
        )
        torch.testing.assert_close(out.cpu().float(), ref, atol=2e-2, rtol=2e-2)


class TestPrefillAttention:
    @pytest.mark.parametrize(
        "Hq,Hkv,q_lens,ctx_lens",
        [
            (32, 8, [64], [0]),               # pure self-attn, aligned
            (32, 8, [33, 7], [0, 0]),         # ragged
            (32, 8, [40, 100], [160, 23]),    # chunked prefill w/ prefix
            (8, 8, [65], [31]),               # G=1
            (8, 1, [29], [11]),               # G=8
        ],
    )
    def test_prefill(self, Hq, Hkv, q_lens, ctx_lens):
        self._run_prefill(Hq, Hkv, q_lens, ctx_lens, 128)

    def test_prefill_d64(self):
        # GPT-2 head_dim: D=64 template instantiation
        self._run_prefill(12, 12, [50], [30], 64)

    def _run_prefill(self, Hq, Hkv, q_lens, ctx_lens, D):
        P = len(q_lens)
        kv_lens = [q + c for q, c in zip(q_lens, ctx_lens)]
        Tq = sum(q_lens)
        q = bf(torch.randn(Tq, Hq, D)).to(DEV)
        kc, vc, bt = make_paged_kv(P, Hkv, D, kv_lens)
        kvl = torch.tensor(kv_lens, dtype=torch.int32, device=DEV)
        cu_q = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(q_lens), 0)), dtype=torch.int32
        ).to(DEV)
        q_pos = torch.cat(
            [
                torch.arange(c, c + ql, dtype=torch.long)
                for ql, c in zip(q_lens, ctx_lens)
            ]
        ).to(DEV)
        out = torch.empty_like(q)
        scale = 1.0 / math.sqrt(D)
        ext.attn_prefill_paged(out, q, cu_q, q_pos, kc, vc, bt, kvl, scale)
        ref = torch_ref.attn_prefill_paged(
            q.cpu().float(), cu_q.cpu(), q_pos.cpu(), kc.cpu().float(),
            vc.cpu().float(), bt.cpu(), kvl.cpu(), scale,
        )
        torch.testing.assert_close(out.cpu().float(), ref, atol=2.5e-2, rtol=2.5e-2)


class TestGemv:
    @pytest.mark.parametrize("M,K,N", [(1, 4096, 4096), (4, 4096, 14336),
                                       (8, 14336, 4096), (3, 4096, 128256)])
    def test_gemv_matches_dense(self, M, K, N):
        x = bf(torch.randn(M, K)).to(DEV)
        w = bf(torch.randn(N, K) / math.sqrt(K)).to(DEV)
        out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
        ext.gemv_bf16(out, x, w)
        ref = (x.cpu().float() @ w.cpu().float().T)
        torch.testing.assert_close(out.cpu().float(), ref, atol=3e-2, rtol=3e-2)

    def test_gemv_strided_x(self):
        M, K, N = 4, 4096, 512
        big = bf(torch.randn(M, 2 * K)).to(DEV)
        x = big[:, :K]  # strided rows
        w = bf(torch.randn(N, K) / math.sqrt(K)).to(DEV)
        out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
        ext.gemv_bf16(out, x, w)
        ref = x.cpu().float() @ w.cpu().float().T
        torch.testing.assert_close(out.cpu().float(), ref, atol=3e-2, rtol=3e-2)


class TestMoeGrouped:
    def test_grouped_matches_reference(self):
        from dts_amd.ops import torch_ref

        E, N, K, P = 4, 512, 1024, 13
        x = bf(torch.randn(P, K)).to(DEV)
        w = bf(torch.randn(E, N, K) / math.sqrt(K)).to(DEV)
        counts = torch.tensor([3, 0, 6, 4], dtype=torch.int32)
        offsets = torch.tensor([0, 3, 3, 9], dtype=torch.int32)
        out = torch.empty(P, N, dtype=torch.bfloat16, device=DEV)
        ext.moe_grouped_linear(out, x, w, counts.to(DEV), offsets.to(DEV))
        ref = torch_ref.moe_grouped_linear(
            x.cpu(), w.cpu(), counts, offsets
        )
        torch.testing.assert_close(
            out.cpu().float(), ref.float(), atol=3e-2, rtol=3e-2
        )

    def test_segment_longer_than_16(self):
        from dts_amd.ops import torch_ref

        E, N, K, P = 2, 256, 512, 40
        x = bf(torch.randn(P, K)).to(DEV)
        w = bf(torch.randn(E, N, K) / math.sqrt(K)).to(DEV)
        counts = torch.tensor([25, 15], dtype=torch.int32)
        offsets = torch.tensor([0, 25], dtype=torch.int32)
        out = torch.empty(P, N, dtype=torch.bfloat16, device=DEV)
        ext.moe_grouped_linear(out, x, w, counts.to(DEV), offsets.to(DEV))
        ref = torch_ref.moe_grouped_linear(x.cpu(), w.cpu(), counts, offsets)
        torch.testing.assert_close(
            out.cpu().float(), ref.float(), atol=3e-2, rtol=3e-2
        )


class TestGemmSkinny:
    @pytest.mark.parametrize(
        "M,K,N",
        [
            (3, 4096, 4096),
            (6, 4096, 14336),
            (8, 14336, 4096),
            (16, 4096, 6144),
            (5, 4096, 128256),
            (16, 4096, 100),  # N not a multiple of the 16-wide tile
        ],
    )
    def test_skinny_matches_dense(self, M, K, N):
        x = bf(torch.randn(M, K)).to(DEV)
        w = bf(torch.randn(N, K) / math.sqrt(K)).to(DEV)
        out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
        ext.gemm_skinny_bf16(out, x, w)
        ref = x.cpu().float() @ w.cpu().float().T
        torch.testing.assert_close(out.cpu().float(), ref, atol=3e-2, rtol=3e-2)

    def test_skinny_strided_x_and_out(self):
        M, K, N = 6, 4096, 512
        big = bf(torch.randn(M, 2 * K)).to(DEV)
        x = big[:, :K]  # strided rows (fused qkv views)
        w = bf(torch.randn(N, K) / math.sqrt(K)).to(DEV)
        out_big = torch.empty(M, 2 * N, dtype=torch.bfloat16, device=DEV)
        out = out_big[:, :N]
        ext.gemm_skinny_bf16(out, x, w)
        ref = x.cpu().float() @ w.cpu().float().T
        torch.testing.assert_close(out.cpu().float(), ref, atol=3e-2, rtol=3e-2)


class TestSampling:
    def test_greedy_matches_argmax(self):
        S, V = 7, 128256
        logits = torch.randn(S, V, device=DEV)
        out = torch.empty(S, dtype=torch.long, device=DEV)
        ext.top_p_sample(
            out,
            logits,
            torch.zeros(S, device=DEV),
            torch.ones(S, device=DEV),
            torch.arange(S, dtype=torch.long, device=DEV),
        )
        assert torch.equal(out.cpu(), logits.argmax(dim=-1).cpu())

    def test_top_p_mass_constraint(self):
        """Sampled tokens must lie in the top-p nucleus."""
        S, V = 16, 128256
        logits = torch.randn(S, V, device=DEV) * 3
        temps = torch.full((S,), 0.7, device=DEV)
        tps = torch.full((S,), 0.9, device=DEV)
        out = torch.empty(S, dtype=torch.long, device=DEV)
        ext.top_p_sample(out, logits, temps, tps, torch.arange(S, dtype=torch.long, device=DEV))
        probs = torch.softmax(logits.float() / 0.7, dim=-1)
        for i in range(S):
            sp, si = torch.sort(probs[i], descending=True)
            cum = torch.cumsum(sp, 0)
            # nucleus: tokens whose exclusive-cumulative < p (+ small slack
            # for the bisection threshold granularity)
            k = int((cum - sp < 0.9).sum())
            nucleus = set(si[: k + 32].tolist())
            assert int(out[i]) in nucleus

    def _v2(self, out, logits, temps, tps, seeds):
        S = logits.shape[0]
        ws = (
            torch.zeros(S, dtype=torch.int64, device=DEV),
            torch.zeros(S, 1024, dtype=torch.float32, device=DEV),
            torch.zeros(S, 16, dtype=torch.float32, device=DEV),
            torch.zeros(S, dtype=torch.float32, device=DEV),
        )
        ext.top_p_sample_v2(out, logits, temps, tps, seeds, *ws)

    def test_v2_greedy_matches_argmax(self):
        S, V = 7, 128256
        logits = torch.randn(S, V, device=DEV)
        out = torch.empty(S, dtype=torch.long, device=DEV)
        self._v2(
            out, logits,
            torch.zeros(S, device=DEV), torch.ones(S, device=DEV),
            torch.arange(S, dtype=torch.long, device=DEV),
        )
        assert torch.equal(out.cpu(), logits.argmax(dim=-1).cpu())

    def test_v2_greedy_tie_breaks_lowest(self):
        V = 8192
        logits = torch.full((1, V), -5.0, device=DEV)
        logits[0, 137] = 3.0
        logits[0, 4242] = 3.0  # exact tie: lowest index must win
        out = torch.empty(1, dtype=torch.long, device=DEV)
        self._v2(
            out, logits,
            torch.zeros(1, device=DEV), torch.ones(1, device=DEV),
            torch.zeros(1, dtype=torch.long, device=DEV),
        )
        assert int(out[0]) == 137

    def test_v2_top_p_mass_constraint(self):
        S, V = 16, 128256
        logits = torch.randn(S, V, device=DEV) * 3
        temps = torch.full((S,), 0.7, device=DEV)
        tps = torch.full((S,), 0.9, device=DEV)
        out = torch.empty(S, dtype=torch.long, device=DEV)
        self._v2(out, logits, temps, tps,
                 torch.arange(S, dtype=torch.long, device=DEV))
        probs = torch.softmax(logits.float() / 0.7, dim=-1)
        for i in range(S):
            sp, si = torch.sort(probs[i], descending=True)
            cum = torch.cumsum(sp, 0)
            k = int((cum - sp < 0.9).sum())
            nucleus = set(si[: k + 32].tolist())
            assert int(out[i]) in nucleus

    def test_v2_deterministic(self):
        S, V = 4, 128256
        logits = torch.randn(S, V, device=DEV)
        temps = torch.full((S,), 0.7, device=DEV)
        tps = torch.full((S,), 0.95, device=DEV)
        seeds = torch.tensor([11, 22, 33, 44], dtype=torch.long, device=DEV)
        outs = []
        for _ in range(2):
            out = torch.empty(S, dtype=torch.long, device=DEV)
            self._v2(out, logits, temps, tps, seeds)
            outs.append(out.cpu())
        assert torch.equal(outs[0], outs[1])

    def test_v2_distribution(self):
        """Over many draws, frequencies track the renormalized nucleus."""
        V = 8192
        logits = torch.randn(1, V, device=DEV) * 2
        temps = torch.full((1,), 1.0, device=DEV)
        tps = torch.full((1,), 0.95, device=DEV)
        counts = torch.zeros(V)
        N = 2000
        out = torch.empty(1, dtype=torch.long, device=DEV)
        for s in range(N):
            self._v2(
                out, logits, temps, tps,
                torch.tensor([s * 7919 + 13], dtype=torch.long, device=DEV),
            )
            counts[int(out[0])] += 1
        probs = torch.softmax(logits[0].float(), dim=-1).cpu()
        top = torch.topk(probs, 5).indices
        for t in top:
            expected = float(probs[t]) / 0.95 * N
            assert abs(counts[t] - expected) < max(6 * math.sqrt(expected), 25)

    def test_sampling_distribution(self):
        """Over many draws, frequencies track the renormalized nucleus."""
        V = 1024
        logits = torch.randn(1, V, device=DEV) * 2
        temps = torch.full((1,), 1.0, device=DEV)
        tps = torch.full((1,), 0.95, device=DEV)
        counts = torch.zeros(V)
        N = 2000
        out = torch.empty(1, dtype=torch.long, device=DEV)
        for s in range(N):
            ext.top_p_sample(
                out, logits, temps, tps,
                torch.tensor([s * 7919 + 13], dtype=torch.long, device=DEV),
            )
            counts[int(out[0])] += 1
        probs = torch.softmax(logits[0].float(), dim=-1).cpu()
        top = torch.topk(probs, 5).indices
        for t in top:
            expected = float(probs[t]) / 0.95 * N
            assert abs(counts[t] - expected) < max(6 * math.sqrt(expected), 25)
