"""GPU engine tests: the full serving path on MI355X with Llama-3-8B
(random-init) — all hot ops through the HIP extension."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from dts_amd.llm.types import SamplingParams
from dts_amd.serving import ServingEngine


@pytest.fixture(scope="module")
def engine():
    eng = ServingEngine(
        model_name="llama-3-8b",
        device="cuda:0",
        dtype=torch.bfloat16,
        kv_memory_bytes=16 << 30,
        weight_seed=0,
    )
    yield eng
    eng.stop()


def _gen(engine, prompt_ids, **kw):
    kw.setdefault("max_tokens", 16)
    kw.setdefault("seed", 0)
    fut = engine.submit_tokens(list(prompt_ids), SamplingParams(**kw))
    engine.run_until_idle()
    return fut.result(timeout=120)


class TestGPUEngine:
    def test_hip_ext_is_loaded(self):
        from dts_amd import ops

        assert ops.hip_available(), "HIP extension must load on GPU"

    def test_generate(self, engine):
        res = _gen(engine, range(1, 40), max_tokens=16)
        assert res.completion_tokens >= 1
        assert res.finish_reason in ("stop", "length")

    def test_greedy_deterministic(self, engine):
        a = _gen(engine, range(1, 100), max_tokens=12, temperature=0.0)
        b = _gen(engine, range(1, 100), max_tokens=12, temperature=0.0)
        assert a.token_ids == b.token_ids

    def test_prefix_cache_consistency(self, engine):
        """Greedy decode must be identical cold vs through the prefix cache."""
        prompt = list(range(7, 700))
        cold = _gen(engine, prompt, max_tokens=12, temperature=0.0)
        hits0 = engine.block_manager.cache_hit_tokens
        warm = _gen(engine, prompt, max_tokens=12, temperature=0.0)
        assert engine.block_manager.cache_hit_tokens - hits0 >= 600
        assert warm.token_ids == cold.token_ids

    def test_concurrent_mixed_batch(self, engine):
        futs = [
            engine.submit_tokens(
                list(range(3, 3 + 50 + 37 * i)),
                SamplingParams(max_tokens=8, seed=i),
            )
            for i in range(6)
        ]
        engine.run_until_idle()
        for f in futs:
            r = f.result(timeout=120)
            assert r.completion_tokens >= 1

    def test_chunked_prefill_matches_single(self, engine):
        """A prompt prefilled in 128-token chunks (separate engine with its
        own cold KV pool, same weights) decodes identically to the one-shot
        prefill on the module engine."""
        prompt = list(range(11, 900))
        big = _gen(engine, prompt, max_tokens=10, temperature=0.0)
        e_small = ServingEngine(
            model_name="llama-3-8b",
            device="cuda:0",
            dtype=torch.bfloat16,
            kv_memory_bytes=4 << 30,
            max_batch_tokens=128,
            model=engine.model,  # share weights — no second init
        )
        small = _gen(e_small, prompt, max_tokens=10, temperature=0.0)
        e_small.stop()
        torch.cuda.empty_cache()
        assert small.token_ids == big.token_ids
