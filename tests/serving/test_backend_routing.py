"""LocalBackend phase-marker → FormGuide routing tests (CPU)."""

import pytest
import torch

from dts_amd.llm.types import Message, SamplingParams
from dts_amd.serving import LocalBackend, ServingEngine
from dts_amd.serving.structured import FormGuide, _RankingGuide


@pytest.fixture(scope="module")
def backend():
    eng = ServingEngine(
        model_name="llama-tiny",
        device="cpu",
        dtype=torch.float32,
        num_blocks=256,
        block_size=8,
        weight_seed=1,
    )
    b = LocalBackend.single(eng, name="llama-tiny")
    yield b
    b.shutdown()


def guide_for(backend, system, user, json_mode=True):
    engine = backend._engine(None)
    msgs = [Message.system(system), Message.user(user)]
    return backend._build_guide(engine, msgs) if json_mode else None


class TestGuideRouting:
    def test_strategy_marker_and_count(self, backend):
        g = guide_for(backend, "[dts:strategy] x", "Propose exactly 5 strategies")
        assert isinstance(g, FormGuide)
        # 5 unique fixed "Strategy i:" key prefixes in the skeleton
        fixed_text = "".join(
            seg.text for seg in g.segments if hasattr(seg, "text")
        )
        assert fixed_text.count("Strategy") == 5

    def test_intent_marker(self, backend):
        g = guide_for(backend, "[dts:intent] x", "Produce exactly 2 distinct user intents")
        fixed_text = "".join(
            seg.text for seg in g.segments if hasattr(seg, "text")
        )
        assert fixed_text.count('"id": "intent_') == 2

    def test_absolute_judge_marker(self, backend):
        g = guide_for(backend, "[dts:judge-absolute] x", "whatever")
        fixed_text = "".join(
            seg.text for seg in g.segments if hasattr(seg, "text")
        )
        assert '"total_score"' in fixed_text
        # criterion scores are Choice segments spanning 0.0-1.0 (full
        # rubric incl. 1.0, ADVICE round-1); only the skeleton is fixed
        assert fixed_text.count('"score": ') == 10  # one per criterion

    def test_comparative_marker_extracts_ids(self, backend):
        ids = [
            "11111111-2222-3333-4444-555555555555",
            "66666666-7777-8888-9999-000000000000",
        ]
        user = "\n".join(f"--- Trajectory {i} (intent: x) ---" for i in ids)
        g = guide_for(backend, "[dts:judge-comparative] x", user)
        assert isinstance(g, _RankingGuide)
        assert g.ids == ids

    def test_unknown_marker_no_guide(self, backend):
        assert guide_for(backend, "[dts:user-sim] x", "y") is None

    def test_non_json_mode_no_guide(self, backend):
        engine = backend._engine(None)
        msgs = [Message.system("[dts:strategy] x"), Message.user("exactly 3")]
        # chat() only builds guides when params.json_mode — mirrored here
        assert backend._build_guide(engine, msgs) is not None  # marker present
        # but the plain-completion path never calls _build_guide (see chat)
