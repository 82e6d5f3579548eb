"""Multi-engine serving: actor on one model, judges on another
(BASELINE.json config 5 shape, tiny models on CPU)."""

import asyncio

import pytest
import torch

from dts_amd.llm import LLM
from dts_amd.search import DTSConfig, DTSEngine
from dts_amd.search.config import GenerationBudget
from dts_amd.serving import LocalBackend, ServingEngine


@pytest.mark.timeout(300)
def test_actor_and_mixtral_judges():
    common = dict(
        device="cpu", dtype=torch.float32, num_blocks=2048, block_size=16,
        max_batch_tokens=2048,
    )
    actor = ServingEngine(model_name="llama-tiny", weight_seed=5, **common)
    judge = ServingEngine(model_name="mixtral-tiny", weight_seed=6, **common)
    backend = LocalBackend(
        {"llama-tiny": actor, "mixtral-tiny": judge}, default_model="llama-tiny"
    )
    llm = LLM(backend, default_model="llama-tiny")
    cfg = DTSConfig(
        goal="g",
        first_message="please help me decide something",
        init_branches=2,
        turns_per_branch=1,
        scoring_mode="absolute",
        prune_threshold=0.0,
        judge_model="mixtral-tiny",
        seed=4,
        budget=GenerationBudget(
            strategy=2048, intent=1024, rephrase=24, user=24, assistant=24,
            judge=4096,
        ),
    )
    result = asyncio.run(DTSEngine(llm, cfg).run(rounds=1))
    branches = [n for n in result.all_nodes if n.strategy is not None]
    assert len(branches) == 2
    assert all(len(n.stats.judge_scores) == 3 for n in branches)
    # judges actually ran on the mixtral engine
    assert judge.tokens_sampled > 0
    assert actor.tokens_sampled > 0
    by_model = result.token_usage["by_model"]
    assert "mixtral-tiny" in by_model and by_model["mixtral-tiny"]["requests"] == 6
    backend.shutdown()
