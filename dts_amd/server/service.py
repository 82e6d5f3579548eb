"""Session service: SearchRequest → engine run → event stream.

Parity: reference backend/services/dts_service.py:17-98 — request→config
mapping, engine lifecycle, asyncio.Queue event bridge with 0.1 s-poll
drain, terminal `complete` event carrying the result + exploration dict
(the tree-state JSON checkpoint). The LLM factory here returns the shared
in-process LocalBackend instead of constructing an HTTP client per
session.
"""

from __future__ import annotations

import asyncio
from typing import AsyncIterator

from dts_amd.llm.backend import LLM
from dts_amd.search import DTSConfig, DTSEngine
from dts_amd.server.schemas import SearchRequest
from dts_amd.utils.logging import logger


def create_dts_config(request: SearchRequest) -> DTSConfig:
    """Map the wire request onto DTSConfig (ref dts_service.py:26-40),
    including the fields the reference dropped."""
    return DTSConfig(
        goal=request.goal,
        first_message=request.first_message,
        init_branches=request.init_branches,
        turns_per_branch=request.turns_per_branch,
        user_intents_per_branch=request.user_intents_per_branch,
        scoring_mode=request.scoring_mode,
        prune_threshold=request.prune_threshold,
        deep_research=request.deep_research,
        strategy_model=request.strategy_model,
        simulator_model=request.simulator_model,
        judge_model=request.judge_model,
        user_variability=request.user_variability,
        reasoning_enabled=request.reasoning_enabled,
        checkpoint_path=request.checkpoint_path,
    )


async def run_dts_session(
    request: SearchRequest, llm: LLM
) -> AsyncIterator[dict]:
    """Run a search, yielding events as they arrive plus a final
    `complete` event (ref dts_service.py:43-98)."""
    config = create_dts_config(request)
    researcher = None
    if config.deep_research:
        from dts_amd.search.researcher import DeepResearcher

        researcher = DeepResearcher(
            llm,
            model=config.strategy_model or config.model,
            cache_dir=config.research_cache_dir,
        )
    engine = DTSEngine(llm, config, researcher=researcher)

    queue: asyncio.Queue = asyncio.Queue()

    async def on_event(event_type: str, data: dict) -> None:
        await queue.put({"type": event_type, "data": data})

    engine.set_event_callback(on_event)
    task = asyncio.create_task(
        engine.run(rounds=request.rounds, resume_from=request.resume_from)
    )

    try:
        while True:
            if task.done() and queue.empty():
                break
            try:
                event = await asyncio.wait_for(queue.get(), timeout=0.1)
                yield event
            except asyncio.TimeoutError:
                continue
        result = await task
        yield {
            "type": "complete",
            "data": {
                "best_node_id": result.best_node_id,
                "best_score": result.best_score,
                "pruned_count": result.pruned_count,
                "total_rounds": result.total_rounds,
                "exploration": result.to_exploration_dict(),
            },
        }
    except Exception as e:  # noqa: BLE001
        logger.exception("search session failed: %s", e)
        yield {"type": "error", "data": {"message": str(e)}}
    finally:
        if not task.done():
            task.cancel()
