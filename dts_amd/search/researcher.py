"""Deep-research component.

Parity: reference backend/core/dts/components/researcher.py:28-285 —
SHA256(goal::first_message)-keyed JSON cache ({"report": ...}, format
kept compatible with the reference's .cache/research/<sha256>.json),
LLM query distillation (ref :241-261) with concatenation fallback, and
the research_log event stream.

The reference shells out to the `gpt_researcher` package, which needs
web access and external API keys; this environment has neither, so the
provider is pluggable: `web` (gpt_researcher, used when the package and
keys exist), or `local` (the serving engine itself writes the briefing —
a long-form generation on the strategy model). Both are cached
identically.
"""

from __future__ import annotations

import asyncio
import hashlib
import json
from pathlib import Path
from typing import Any, Callable, Optional

from dts_amd.llm.backend import LLM
from dts_amd.llm.types import Message
from dts_amd.search import prompts
from dts_amd.search.events import create_event_emitter
from dts_amd.utils.logging import log_phase, logger


class DeepResearcher:
    def __init__(
        self,
        llm: LLM,
        model: Optional[str] = None,
        cache_dir: str = ".cache/research",
        max_concurrent_research: int = 1,
        on_cost: Optional[Callable[[float], None]] = None,
        on_event: Optional[Callable[[str, dict], Any]] = None,
        provider: str = "auto",  # auto | web | local
        local_report_tokens: int = 1024,
    ) -> None:
        self.llm = llm
        self.model = model
        self.cache_dir = Path(cache_dir)
        self.cache_dir.mkdir(parents=True, exist_ok=True)
        self._sem = asyncio.Semaphore(max_concurrent_research)
        self._on_cost = on_cost
        self._emit = create_event_emitter(on_event)
        self.provider = provider
        self.local_report_tokens = local_report_tokens

    # ------------------------------------------------------------------
    async def research(self, goal: str, first_message: str) -> str:
        cache_key = self._cache_key(goal, first_message)
        cached = self._load_cache(cache_key)
        if cached:
            log_phase("RESEARCH", f"Cache hit: {cache_key[:8]}...", indent=1)
            self._emit(
                "research_log",
                {"message": "Using cached research results", "type": "cache_hit"},
            )
            return cached

        self._emit(
            "research_log",
            {"message": "Generating research query...", "type": "progress"},
        )
        query = await self._generate_query(goal, first_message)
        log_phase("RESEARCH", f"Query: {query[:80]}", indent=1)

        provider = self.provider
        if provider == "auto":
            provider = "web" if self._web_available() else "local"

        async with self._sem:
            if provider == "web":
                report = await self._web_research(query)
            else:
                report = await self._local_research(goal, first_message, query)

        self._save_cache(cache_key, report)
        return report

    # ------------------------------------------------------------------
    @staticmethod
    def _web_available() -> bool:
        try:
            import gpt_researcher  # noqa: F401

            return True
        except ImportError:
            return False

    async def _web_research(self, query: str) -> str:
        from gpt_researcher import GPTResearcher  # type: ignore

        self._emit(
            "research_log",
            {"message": f"Researching: {query[:80]}...", "type": "start"},
        )
        researcher = GPTResearcher(query=query, report_type="deep")
        await researcher.conduct_research()
        report = await researcher.write_report()
        cost = researcher.get_costs()
        if self._on_cost and cost:
            self._on_cost(cost)
        return report

    async def _local_research(self, goal: str, first_message: str, query: str) -> str:
        """Briefing written by the local model (no network)."""
        self._emit(
            "research_log",
            {"message": "Writing local briefing (no web access)...", "type": "start"},
        )
        system = (
            "[dts:research-report] Write a concise background briefing that "
            "would help an assistant handle the following conversation well: "
            "key facts, likely user concerns, common pitfalls, and concrete "
            "talking points."
        )
        user = (
            f"Conversation goal: {goal}\nOpening message: {first_message}\n"
            f"Research focus: {query}\n\nBriefing:"
        )
        completion = await self.llm.complete(
            [Message.system(system), Message.user(user)],
            model=self.model,
            temperature=0.3,
            max_tokens=self.local_report_tokens,
        )
        return completion.message.content or ""

    async def _generate_query(self, goal: str, first_message: str) -> str:
        system, user = prompts.research_query_distill(goal, first_message)
        try:
            completion = await self.llm.complete(
                [Message.system(system), Message.user(user)],
                model=self.model,
                temperature=0.3,
                max_tokens=96,
            )
            query = (completion.message.content or "").strip()
            if query:
                return query
        except Exception as e:  # noqa: BLE001
            logger.warning("Query generation failed, using fallback: %s", e)
        return f"{goal} - {first_message}"

    # ------------------------------------------------------------------
    @staticmethod
    def _cache_key(goal: str, first_message: str) -> str:
        return hashlib.sha256(f"{goal}::{first_message}".encode()).hexdigest()

    def _load_cache(self, key: str) -> Optional[str]:
        path = self.cache_dir / f"{key}.json"
        if path.exists():
            try:
                return json.loads(path.read_text()).get("report")
            except (json.JSONDecodeError, OSError):
                return None
        return None

    def _save_cache(self, key: str, report: str) -> None:
        try:
            (self.cache_dir / f"{key}.json").write_text(json.dumps({"report": report}))
        except OSError as e:
            logger.warning("Failed to cache research: %s", e)
