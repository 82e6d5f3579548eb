"""Trajectory judging.

Parity: reference backend/core/dts/components/evaluator.py:21-373 —
absolute mode (3 parallel judges → median vote, critique from the
median-closest judge, ref :160-223), comparative mode (group siblings by
parent, one forced-ranking call per group, synthetic [s,s,s] scores,
absolute fallback on invalid ranking, ref :102-158, 234-347).

MI355X note: a judge call re-reads a whole trajectory the engine just
generated — with shared-prefix paged KV those prompt tokens are already
cached, so judging is mostly a large-batch prefill of the rubric suffix
(SURVEY.md §2.3 judge rows).
"""

from __future__ import annotations

import asyncio
from typing import Any, Callable, Optional

from dts_amd.llm.backend import LLM
from dts_amd.llm.errors import ContextLengthError
from dts_amd.llm.types import Message
from dts_amd.search import prompts
from dts_amd.search.aggregator import aggregate_majority_vote
from dts_amd.search.retry import llm_retry
from dts_amd.search.types import (
    AggregatedScore,
    DialogueNode,
    format_message_history,
)
from dts_amd.utils.logging import log_phase, logger


def _shrink(text: str, frac: float) -> str:
    """Middle-out truncation: keep the opening and the (decisive) tail.

    Local models have a hard context; a judge prompt that cannot fit is
    retried with shrunk trajectories instead of silently scoring 0.0
    (the reference just surfaces the provider's HTTP 400 —
    ref client.py:441-442 — because remote contexts are effectively
    unbounded for chat-scale inputs)."""
    if frac >= 1.0 or len(text) < 256:
        return text
    n = max(128, int(len(text) * frac))
    head = n // 3
    tail = n - head
    return (
        text[:head] + "\n... [conversation truncated to fit context] ...\n" + text[-tail:]
    )


class TrajectoryEvaluator:
    def __init__(
        self,
        llm: LLM,
        goal: str,
        model: Optional[str] = None,
        judge_temperature: float = 0.3,
        prune_threshold: float = 6.5,
        max_concurrency: int = 16,
        on_usage: Optional[Callable[[Any, str], None]] = None,
        deep_research_context: Optional[str] = None,
        max_tokens: int = 1024,
        seed: Optional[int] = None,
        comparative_split: bool = False,
    ) -> None:
        self.llm = llm
        self.goal = goal
        self.model = model
        self.judge_temperature = judge_temperature
        self.prune_threshold = prune_threshold
        self.max_tokens = max_tokens
        self.seed = seed
        self._sem = asyncio.Semaphore(max_concurrency)
        self._on_usage = on_usage
        self.deep_research_context = deep_research_context
        self._judge_counter = 0
        # Sibling groups larger than this are ranked in chunks: a single
        # forced-ranking prompt embeds EVERY sibling trajectory
        # (ref prompts.py:349-355 — the largest prompt in the system), so
        # a 48-branch group (weak-scaling bench at N=8) would need ~100k
        # tokens of context. The reference caps init_branches at 20 and
        # would overflow its providers' context the same way; chunked
        # ranking preserves the forced-ranking semantics within chunks.
        self.max_comparative_group = 8
        # split comparative mode: one parallel critique call per sibling
        # (same full-context prompt — shared KV prefix makes the repeated
        # prefill free) plus one compact ranking-only call. Cuts the
        # score phase's sequential decode depth ~4x vs the single
        # combined guided generation (round-1 VERDICT weak #5).
        self.comparative_split = comparative_split

    def set_research_context(self, context: Optional[str]) -> None:
        self.deep_research_context = context

    # ------------------------------------------------------------------
    async def evaluate_absolute(self, nodes: list) -> dict:
        results = await asyncio.gather(
            *[self._judge_single(node) for node in nodes], return_exceptions=True
        )
        scores: dict = {}
        for node, result in zip(nodes, results):
            if isinstance(result, Exception):
                logger.error("Error judging node %s: %s", node.id, result)
                scores[node.id] = AggregatedScore.zero(self.prune_threshold)
            else:
                agg, critiques = result
                scores[node.id] = agg
                node.update_with_evaluation(agg, critiques)
        return scores

    def comparative_chunks(self, nodes: list) -> list:
        """Deterministic (parent_id, chunk) list: siblings grouped by
        parent, oversized groups split into <=max_comparative_group chunks
        (used by both the local and the DP-sharded paths)."""
        groups: dict = {}
        for node in nodes:
            groups.setdefault(node.parent_id or "root", []).append(node)
        chunks: list = []
        for parent_id in sorted(groups):
            group = sorted(groups[parent_id], key=lambda n: n.id)
            g = self.max_comparative_group
            for i in range(0, len(group), g):
                chunk = group[i : i + g]
                # avoid a trailing singleton when the group splits unevenly
                if len(chunk) == 1 and chunks and chunks[-1][0] == parent_id:
                    chunks[-1][1].append(chunk[0])
                else:
                    chunks.append((parent_id, chunk))
        return chunks

    async def evaluate_comparative(self, nodes: list) -> dict:
        if len(nodes) <= 1:
            return await self.evaluate_absolute(nodes)

        tasks = []
        for parent_id, group in self.comparative_chunks(nodes):
            if len(group) == 1:
                tasks.append(self._judge_single_wrapped(group[0]))
            else:
                tasks.append(self._judge_group_comparative(parent_id, group))

        results = await asyncio.gather(*tasks, return_exceptions=True)
        scores: dict = {}
        for result in results:
            if isinstance(result, Exception):
                logger.error("Judge task failed: %s", result)
                continue
            if isinstance(result, dict):
                scores.update(result)
        return scores

    # ------------------------------------------------------------------
    async def _judge_single(self, node: DialogueNode):
        history_str = format_message_history(node.messages)
        results: list = []
        for frac in (1.0, 0.5, 0.25):
            system, user = prompts.trajectory_outcome_judge(
                conversation_goal=self.goal,
                conversation_history=_shrink(history_str, frac),
                deep_research_context=self.deep_research_context,
            )
            results = await asyncio.gather(
                *[self._call_json(system, user) for _ in range(3)],
                return_exceptions=True,
            )
            if not all(isinstance(r, ContextLengthError) for r in results):
                break
            logger.warning(
                "Judge prompt exceeds context; retrying at %.0f%% history",
                frac * 50,
            )

        scores: list = []
        judge_results: list = []
        for result in results:
            if isinstance(result, dict) and "total_score" in result:
                try:
                    scores.append(float(result["total_score"]))
                    judge_results.append(result)
                    continue
                except (TypeError, ValueError):
                    pass
            if isinstance(result, Exception):
                logger.warning("Judge failed: %s", result)
            scores.append(0.0)
            judge_results.append({})

        agg = aggregate_majority_vote(scores[:3], pass_threshold=self.prune_threshold)

        # critique from the judge closest to the median (ref evaluator.py:195-221)
        critiques = None
        closest = min(range(3), key=lambda i: abs(scores[i] - agg.aggregated_score))
        median_result = judge_results[closest]
        if median_result:
            strengths, weaknesses = [], []
            for name, data in (median_result.get("criteria") or {}).items():
                if not isinstance(data, dict):
                    continue
                score = data.get("score", 1.0)
                rationale = data.get("rationale", "")
                if isinstance(score, (int, float)) and rationale:
                    if score < 0.5:
                        weaknesses.append(f"{name}: {rationale}")
                    elif score >= 0.8:
                        strengths.append(f"{name}: {rationale}")
            critiques = {
                "strengths": strengths,
                "weaknesses": weaknesses,
                "key_moment": median_result.get("key_turning_point"),
                "summary": median_result.get("summary"),
                "biggest_missed_opportunity": median_result.get(
                    "biggest_missed_opportunity"
                ),
            }
        return agg, critiques

    async def _judge_single_wrapped(self, node: DialogueNode) -> dict:
        agg, critiques = await self._judge_single(node)
        node.update_with_evaluation(agg, critiques)
        return {node.id: agg}

    async def _judge_group_comparative(self, parent_id: str, group: list) -> dict:
        log_phase("JUDGE", f"Ranking {len(group)} siblings...", indent=1)
        # the group prompt embeds EVERY sibling trajectory — on a small
        # context it is the first thing to overflow, so retry the whole
        # group judgment at shrinking history fractions before giving up
        for frac in (1.0, 0.5, 0.25):
            trajectories = [
                {
                    "id": node.id,
                    "intent_label": (
                        node.user_intent.label if node.user_intent else "unknown"
                    ),
                    "history": _shrink(format_message_history(node.messages), frac),
                }
                for node in group
            ]
            try:
                if self.comparative_split and len(group) > 1:
                    return await self._judge_group_split(group, trajectories)
                system, user = prompts.comparative_trajectory_judge(
                    conversation_goal=self.goal,
                    trajectories=trajectories,
                    deep_research_context=self.deep_research_context,
                )
                try:
                    result = await self._call_json(system, user)
                except ContextLengthError:
                    raise
                except Exception as e:  # noqa: BLE001
                    logger.warning(
                        "Comparative judge errored (%s); absolute fallback", e
                    )
                    result = None
                if not result or "ranking" not in result:
                    return await self._fallback_absolute(group)
                return self._apply_ranking(group, result)
            except ContextLengthError:
                logger.warning(
                    "Group ranking prompt exceeds context; retrying at "
                    "%.0f%% history",
                    frac * 50,
                )
                continue
        return await self._fallback_absolute(group)

    async def _judge_group_split(self, group: list, trajectories: list) -> dict:
        """Split comparative judging: n parallel critique calls + one
        ranking-only call, all sharing the (goal + trajectories) prompt
        prefix so only one real prefill hits the engine."""
        crit_prompts = [
            prompts.comparative_critique_judge(
                conversation_goal=self.goal,
                trajectories=trajectories,
                target_id=node.id,
                deep_research_context=self.deep_research_context,
            )
            for node in group
        ]
        rank_prompt = prompts.comparative_ranking_judge(
            conversation_goal=self.goal,
            trajectories=trajectories,
            deep_research_context=self.deep_research_context,
        )
        results = await asyncio.gather(
            *[self._call_json(s, u) for s, u in crit_prompts],
            self._call_json(*rank_prompt),
            return_exceptions=True,
        )
        rank_res = results[-1]
        if isinstance(rank_res, ContextLengthError):
            raise rank_res  # caller retries the group at a smaller frac
        if (
            isinstance(rank_res, Exception)
            or not isinstance(rank_res, dict)
            or "ranking" not in rank_res
        ):
            logger.warning("Split ranking failed; absolute fallback")
            return await self._fallback_absolute(group)
        critiques: dict = {}
        for node, r in zip(group, results):
            if isinstance(r, dict):
                critiques[node.id] = r
        merged = {
            "ranking": rank_res.get("ranking"),
            "critiques": critiques,
            "ranking_confidence": rank_res.get("ranking_confidence"),
        }
        return self._apply_ranking(group, merged)

    def _apply_ranking(self, group: list, result: dict) -> dict:
        scores: dict = {}
        critiques = result.get("critiques", {}) or {}
        for entry in result.get("ranking", []):
            if not isinstance(entry, dict):
                continue
            node_id = entry.get("trajectory_id", "")
            node = next((n for n in group if n.id == node_id), None)
            if node is None:
                continue
            try:
                score = float(entry.get("score", 0.0))
            except (TypeError, ValueError):
                score = 0.0
            agg = AggregatedScore(
                individual_scores=[score, score, score],
                aggregated_score=score,
                pass_threshold=self.prune_threshold,
                pass_votes=3 if score >= self.prune_threshold else 0,
                passed=score >= self.prune_threshold,
            )
            scores[node_id] = agg
            node.stats.judge_scores = [score]
            node.stats.aggregated_score = score
            if node_id in critiques:
                node.stats.critiques = critiques[node_id]

        for node in group:  # nodes the ranking omitted (ref evaluator.py:314-319)
            if node.id not in scores:
                scores[node.id] = AggregatedScore.zero(self.prune_threshold)
                node.stats.judge_scores = [0.0]
                node.stats.aggregated_score = 0.0
        return scores

    async def _fallback_absolute(self, group: list) -> dict:
        results = await asyncio.gather(
            *[self._judge_single(node) for node in group], return_exceptions=True
        )
        scores: dict = {}
        for node, result in zip(group, results):
            if isinstance(result, Exception):
                agg, critiques = AggregatedScore.zero(self.prune_threshold), None
            else:
                agg, critiques = result
            scores[node.id] = agg
            node.update_with_evaluation(agg, critiques)
        return scores

    # ------------------------------------------------------------------
    async def _call_json(self, system: str, user: str) -> Optional[dict]:
        async with self._sem:
            return await self._call_json_inner(system, user)

    @llm_retry(max_attempts=3)
    async def _call_json_inner(self, system: str, user: str) -> Optional[dict]:
        seed = None
        if self.seed is not None:
            self._judge_counter += 1
            from dts_amd.utils.seeding import stable_seed

            seed = stable_seed(self.seed, "judge", self._judge_counter)
        completion = await self.llm.complete(
            [Message.system(system), Message.user(user)],
            model=self.model,
            temperature=self.judge_temperature,
            structured_output=True,
            max_tokens=self.max_tokens,
            seed=seed,
        )
        if self._on_usage:
            self._on_usage(completion, "judge")
        return completion.data
