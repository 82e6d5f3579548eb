"""Deterministic fake backends for CPU tests.

The reference's test strategy is mock-everything (MagicMock
LLMs with canned Completions). These fakes are the equivalent seam for this
framework: `FakeBackend` understands the framework's own prompt phases and
produces schema-valid outputs deterministically (hash-seeded), so the entire
search layer runs end-to-end with no GPU and no network; `ScriptedBackend`
replays an explicit list of completions (closest analogue of the reference's
AsyncMock side_effect lists, ref tests/conftest.py:49-55).
"""

from __future__ import annotations

import asyncio
import hashlib
import json
from typing import Optional

from dts_amd.llm.types import Completion, Message, SamplingParams, Usage


def _stable_hash(text: str) -> int:
    return int.from_bytes(hashlib.sha256(text.encode()).digest()[:8], "big")


def _approx_tokens(text: str) -> int:
    return max(1, len(text) // 4)


class FakeBackend:
    """Phase-aware deterministic backend.

    Recognizes the framework's prompt phases by markers emitted by
    dts_amd.search.prompts (each template embeds a stable marker string) and
    returns valid outputs for each: strategy JSON, intent JSON, plain user /
    assistant turns, absolute-judge JSON, comparative-ranking JSON.
    """

    def __init__(self, latency_s: float = 0.0, score_salt: str = "") -> None:
        self.latency_s = latency_s
        self.score_salt = score_salt
        self.calls: list[dict] = []

    async def chat(
        self,
        messages: list[Message],
        params: SamplingParams,
        model: Optional[str] = None,
    ) -> Completion:
        if self.latency_s:
            await asyncio.sleep(self.latency_s)
        system = messages[0].content if messages and messages[0].role == "system" else ""
        last_user = next(
            (m.content for m in reversed(messages) if m.role == "user"), ""
        )
        self.calls.append({"system": system, "model": model, "n_messages": len(messages)})
        text = self._respond(system or "", last_user or "", messages)
        prompt_text = "\n".join(m.content or "" for m in messages)
        return Completion(
            message=Message.assistant(text),
            usage=Usage(
                prompt_tokens=_approx_tokens(prompt_text),
                completion_tokens=_approx_tokens(text),
            ),
            model=model or "fake",
            finish_reason="stop",
        )

    # ------------------------------------------------------------------
    def _respond(self, system: str, user: str, messages: list[Message]) -> str:
        if "[dts:strategy]" in system:
            return self._strategies(user)
        if "[dts:intent]" in system:
            return self._intents(user)
        if "[dts:rephrase]" in system:
            return f"(rephrased) {user[-200:]}"
        if "[dts:user-sim]" in system:
            h = _stable_hash(user + str(len(messages)))
            return f"That makes sense, but can you elaborate on point {h % 7}?"
        if "[dts:assistant]" in system:
            h = _stable_hash(user + str(len(messages)))
            return f"Certainly — here is a deeper explanation, aspect {h % 11}."
        if "[dts:judge-absolute]" in system:
            return self._absolute_judgment(user)
        if "[dts:judge-comparative]" in system:
            return self._comparative_judgment(user)
        return "Understood."

    def _strategies(self, user: str) -> str:
        import re

        m = re.search(r"exactly (\d+)", user)
        n = int(m.group(1)) if m else 4
        nodes = {
            f"Strategy {i + 1}": f"Deterministic fake strategy number {i + 1}."
            for i in range(n)
        }
        return json.dumps({"goal": "fake", "nodes": nodes, "coverage_rationale": "fake"})

    def _intents(self, user: str) -> str:
        import re

        m = re.search(r"exactly (\d+)", user)
        n = int(m.group(1)) if m else 3
        tones = ["engaged", "skeptical", "confused", "enthusiastic", "anxious"]
        stances = ["accepting", "questioning", "challenging", "exploring"]
        intents = [
            {
                "id": f"intent_{i}",
                "label": f"Intent {i + 1}",
                "description": f"Fake intent {i + 1}.",
                "emotional_tone": tones[i % len(tones)],
                "cognitive_stance": stances[i % len(stances)],
            }
            for i in range(n)
        ]
        return json.dumps({"intents": intents})

    def _score_for(self, text: str) -> float:
        h = _stable_hash(self.score_salt + text)
        return round(3.0 + (h % 700) / 100.0, 1)  # 3.0 .. 9.9

    def _absolute_judgment(self, user: str) -> str:
        total = self._score_for(user)
        per = round(total / 10.0, 2)
        crit_names = [
            "goal_achieved",
            "user_need_addressed",
            "forward_progress",
            "user_engagement_maintained",
            "rapport_preserved",
            "appropriate_resolution",
            "actionable_outcome",
            "no_harm_done",
            "efficient_path",
            "user_better_off",
        ]
        criteria = {
            name: {"score": per, "rationale": f"fake rationale for {name}"}
            for name in crit_names
        }
        return json.dumps(
            {
                "criteria": criteria,
                "total_score": total,
                "confidence": "medium",
                "summary": "Deterministic fake judgment.",
                "key_turning_point": "turn 1",
                "biggest_missed_opportunity": "none",
            }
        )

    def _comparative_judgment(self, user: str) -> str:
        import re

        # rank by trajectory CONTENT (uuid only as tiebreak): node ids are
        # random per run, so an id-hash ranking made seeded searches
        # non-reproducible run to run (caught by the latency-invariance
        # test — a content-blind judge is also just a worse fake)
        parts = re.split(r"--- Trajectory ([0-9a-f-]+)", user)
        ids = parts[1::2]
        bodies = parts[2::2]
        content = dict(zip(ids, bodies))
        order = sorted(
            ids,
            key=lambda i: (
                _stable_hash(self.score_salt + content.get(i, "")),
                i,
            ),
            reverse=True,
        )
        ranking = []
        critiques = {}
        for rank, tid in enumerate(order, start=1):
            score = max(0.0, 7.5 - 1.5 * (rank - 1))
            ranking.append(
                {
                    "rank": rank,
                    "trajectory_id": tid,
                    "score": score,
                    "reason": f"fake rank {rank}",
                }
            )
            critiques[tid] = {
                "weaknesses": ["fake weakness"],
                "strengths": ["fake strength"],
                "key_moment": "turn 1",
            }
        return json.dumps(
            {"critiques": critiques, "ranking": ranking, "ranking_confidence": "medium"}
        )


class ScriptedBackend:
    """Replays an explicit sequence of responses (str or Completion)."""

    def __init__(self, responses: list) -> None:
        self._responses = list(responses)
        self.calls: list[list[Message]] = []

    async def chat(
        self,
        messages: list[Message],
        params: SamplingParams,
        model: Optional[str] = None,
    ) -> Completion:
        self.calls.append(messages)
        if not self._responses:
            raise RuntimeError("ScriptedBackend out of responses")
        r = self._responses.pop(0)
        if isinstance(r, Exception):
            raise r
        if isinstance(r, Completion):
            return r
        return Completion(
            message=Message.assistant(str(r)),
            usage=Usage(prompt_tokens=10, completion_tokens=10),
            model=model or "scripted",
            finish_reason="stop",
        )
