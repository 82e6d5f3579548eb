"""Strategy and user-intent generation.

Parity: reference backend/core/dts/components/generator.py:30-180 —
`generate_strategies` (one structured call → N Strategy), `generate_intents`
(one structured call → K UserIntent), FIXED_INTENT fallback persona
(ref :21-27), bounded concurrency, retry on transient errors.
"""

from __future__ import annotations

import asyncio
from typing import Any, Callable, Optional

from dts_amd.llm.backend import LLM
from dts_amd.llm.types import Message
from dts_amd.search import prompts
from dts_amd.search.retry import llm_retry
from dts_amd.search.types import Strategy, UserIntent, format_message_history
from dts_amd.utils.logging import logger

FIXED_INTENT = UserIntent(
    id="fixed_engaged_critic",
    label="Engaged Critic",
    description="A thoughtful user who engages constructively while maintaining healthy skepticism",
    emotional_tone="curious but skeptical",
    cognitive_stance="analytical, asks probing questions",
)

# per-call lenses for split strategy generation (also used per-rank by
# the DP sharded init, search/dist_engine.py): each call covers a
# distinct angle so the union stays diverse without a single serial
# mega-form
DIVERSITY_ANGLES = [
    "empathy-first emotional connection",
    "pragmatic step-by-step problem solving",
    "data-driven evidence and benchmarks",
    "narrative framing and storytelling",
    "expert-authority framing with credible sourcing",
    "collaborative co-design with the user",
    "contrarian assumption-challenging",
    "incremental trust-building and small commitments",
]


class StrategyGenerator:
    def __init__(
        self,
        llm: LLM,
        goal: str,
        model: Optional[str] = None,
        temperature: float = 0.7,
        max_concurrency: int = 16,
        on_usage: Optional[Callable[[Any, str], None]] = None,
        max_tokens: int = 1024,
        intent_max_tokens: Optional[int] = None,
        seed: Optional[int] = None,
        strategy_split: bool = False,
    ) -> None:
        self.llm = llm
        self.goal = goal
        self.model = model
        self.temperature = temperature
        self.max_tokens = max_tokens
        self.intent_max_tokens = intent_max_tokens or max_tokens
        self.seed = seed
        # split strategy generation: one parallel call per strategy,
        # distinct diversity lens each, shared prompt prefix — cuts the
        # init phase's sequential decode depth ~count x on the local
        # engine. False reproduces the reference's single N-node call.
        self.strategy_split = strategy_split
        # rotated by the DP sharded init so ranks' lens sets don't overlap
        self.lens_offset = 0
        self._sem = asyncio.Semaphore(max_concurrency)
        self._on_usage = on_usage

    async def generate_strategies(
        self,
        first_message: str,
        count: int,
        deep_research_context: Optional[str] = None,
    ) -> list:
        if self.strategy_split and count > 1:
            return await self._generate_strategies_split(
                first_message, count, deep_research_context
            )
        system, user = prompts.conversation_tree_generator(
            num_nodes=count,
            conversation_goal=self.goal,
            conversation_context=first_message,
            deep_research_context=deep_research_context,
        )
        result = await self._call_json(system, user, phase="strategy")
        if not result:
            raise RuntimeError("Strategy generation failed after retries")
        strategies = [
            Strategy(tagline=tagline, description=str(desc))
            for tagline, desc in result.get("nodes", {}).items()
        ]
        return strategies

    async def _generate_strategies_split(
        self,
        first_message: str,
        count: int,
        deep_research_context: Optional[str],
    ) -> list:
        calls = []
        for i in range(count):
            system, user = prompts.conversation_tree_generator_single(
                index=i + 1,
                total=count,
                conversation_goal=self.goal,
                conversation_context=first_message,
                lens=DIVERSITY_ANGLES[
                    (self.lens_offset + i) % len(DIVERSITY_ANGLES)
                ],
                deep_research_context=deep_research_context,
            )
            calls.append(self._call_json(system, user, phase="strategy"))
        results = await asyncio.gather(*calls, return_exceptions=True)
        strategies: list = []
        for i, result in enumerate(results):
            if isinstance(result, Exception) or not isinstance(result, dict):
                logger.warning("Split strategy call %d failed: %s", i + 1, result)
                continue
            for tagline, desc in (result.get("nodes") or {}).items():
                strategies.append(Strategy(tagline=tagline, description=str(desc)))
        if not strategies:
            raise RuntimeError("Strategy generation failed after retries")
        return strategies

    async def generate_intents(self, history: list, count: int) -> list:
        system, user = prompts.user_intent_generator(
            num_intents=count,
            conversation_goal=self.goal,
            conversation_history=format_message_history(history),
        )
        result = await self._call_json(
            system, user, phase="intent", max_tokens=self.intent_max_tokens
        )
        if not result:
            raise RuntimeError("Intent generation failed after retries")
        intents = []
        for data in result.get("intents", []):
            try:
                intents.append(
                    UserIntent(
                        id=str(data.get("id", "unknown")),
                        label=str(data.get("label", "Unknown")),
                        description=str(data.get("description", "")),
                        emotional_tone=str(data.get("emotional_tone", "neutral")),
                        cognitive_stance=str(data.get("cognitive_stance", "neutral")),
                    )
                )
            except Exception as e:  # noqa: BLE001
                logger.warning("Failed to parse intent: %s", e)
        return intents

    async def _call_json(
        self, system: str, user: str, phase: str, max_tokens: Optional[int] = None
    ) -> Optional[dict]:
        async with self._sem:
            return await self._call_json_inner(system, user, phase, max_tokens)

    @llm_retry(max_attempts=3)
    async def _call_json_inner(
        self, system: str, user: str, phase: str, max_tokens: Optional[int] = None
    ) -> Optional[dict]:
        completion = await self.llm.complete(
            [Message.system(system), Message.user(user)],
            model=self.model,
            temperature=self.temperature,
            structured_output=True,
            max_tokens=max_tokens or self.max_tokens,
            seed=self.seed,
        )
        if self._on_usage:
            self._on_usage(completion, phase)
        return completion.data
