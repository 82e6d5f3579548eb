"""Headless example runner (parity: reference main.py:40-61).

Runs a small search on the local serving engine and saves the
tree-state JSON checkpoint to dts_output.json. Uses the tiny model on
CPU and Llama-3-8B on a GPU.
"""

from __future__ import annotations

import asyncio
import time

import torch

from dts_amd.llm import LLM
from dts_amd.search import DTSConfig, DTSEngine
from dts_amd.search.researcher import DeepResearcher
from dts_amd.serving import LocalBackend, ServingEngine


async def run_dts_example(
    *,
    init_branches: int = 3,
    turns_per_branch: int = 2,
    user_intents_per_branch: int = 2,
    rounds: int = 2,
    deep_research: bool = True,
    output_path: str = "dts_output.json",
    resume_from: str | None = None,
    checkpoint_path: str | None = None,
) -> float:
    use_gpu = torch.cuda.is_available()
    model = "llama-3-8b" if use_gpu else "llama-tiny"
    engine_kwargs = {} if use_gpu else {
        "num_blocks": 4096,
        "block_size": 16,
        "dtype": torch.float32,
    }
    serving = ServingEngine(model_name=model, **engine_kwargs)
    backend = LocalBackend.single(serving, name=model)
    llm = LLM(backend, default_model=model)

    config = DTSConfig(
        goal="Convince a skeptical team lead to adopt automated testing",
        first_message="Our team doesn't write tests. Is it really worth the time?",
        init_branches=init_branches,
        turns_per_branch=turns_per_branch,
        user_intents_per_branch=user_intents_per_branch,
        user_variability=True,
        scoring_mode="comparative",
        prune_threshold=5.0,
        deep_research=deep_research,
        checkpoint_path=checkpoint_path,
        seed=0,
    )
    researcher = DeepResearcher(llm, cache_dir=config.research_cache_dir)
    engine = DTSEngine(llm, config, researcher=researcher)
    result = await engine.run(rounds=rounds, resume_from=resume_from)
    result.save_json(output_path)
    print(f"Best score: {result.best_score:.1f} — saved to {output_path}")
    backend.shutdown()
    return result.best_score


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=2)
    ap.add_argument("--resume", default=None, help="exploration JSON to resume")
    ap.add_argument("--checkpoint", default=None,
                    help="write the tree state here after every round")
    a = ap.parse_args()
    start = time.time()
    asyncio.run(run_dts_example(rounds=a.rounds, resume_from=a.resume,
                                checkpoint_path=a.checkpoint))
    print(f"Completed in {time.time() - start:.1f}s")
