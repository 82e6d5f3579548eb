"""Core data types for the search layer.

Parity: reference backend/core/dts/types.py (Strategy :307, UserIntent
:314-323, NodeStatus :298-304, AggregatedScore :352-370, NodeStats
:373-384, DialogueNode :387-428, DTSRunResult :439-563, TokenTracker
:118-295). Differences by design:

 - dataclasses instead of pydantic (no wire validation needed below the
   API layer; the server re-validates at the edge — dts_amd/server/schemas.py);
 - no USD costing: there is no remote API. TokenTracker keeps the same
   per-phase / per-model token books and to_dict() shape with cost fields
   pinned to 0.0, so the exploration-JSON contract stays readable by
   reference-compatible consumers (ref types.py:190-219).
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from enum import Enum
from pathlib import Path
from typing import Optional

from dts_amd.llm.types import Usage


class NodeStatus(str, Enum):
    ACTIVE = "active"
    PRUNED = "pruned"
    TERMINAL = "terminal"
    ERROR = "error"


@dataclass
class Strategy:
    tagline: str
    description: str


@dataclass
class UserIntent:
    id: str
    label: str
    description: str
    emotional_tone: str
    cognitive_stance: str


@dataclass
class AggregatedScore:
    """Median-vote result over exactly 3 judges (ref types.py:352-370)."""

    individual_scores: list
    aggregated_score: float
    pass_threshold: float = 5.0
    pass_votes: int = 0
    passed: bool = False

    @classmethod
    def zero(cls, threshold: float = 5.0) -> "AggregatedScore":
        return cls(
            individual_scores=[0.0, 0.0, 0.0],
            aggregated_score=0.0,
            pass_threshold=threshold,
            pass_votes=0,
            passed=False,
        )


@dataclass
class NodeStats:
    visits: int = 0
    value_sum: float = 0.0
    value_mean: float = 0.0
    judge_scores: list = field(default_factory=list)
    aggregated_score: float = 0.0
    critiques: dict = field(default_factory=dict)


@dataclass
class DialogueNode:
    """A conversation state in the tree (ref types.py:387-428)."""

    id: str
    parent_id: Optional[str] = None
    children: list = field(default_factory=list)
    depth: int = 0
    status: NodeStatus = NodeStatus.ACTIVE
    strategy: Optional[Strategy] = None
    user_intent: Optional[UserIntent] = None
    messages: list = field(default_factory=list)
    stats: NodeStats = field(default_factory=NodeStats)
    prune_reason: Optional[str] = None
    # MI355X addition: the serving-engine sequence whose paged KV blocks
    # hold this node's conversation prefix, enabling fork-time KV sharing
    # (SURVEY.md §2.3 "shared-prefix refcounting").
    kv_session: Optional[object] = None

    @property
    def strategy_label(self) -> str:
        return self.strategy.tagline if self.strategy else "unknown"

    @property
    def intent_label(self) -> Optional[str]:
        return self.user_intent.label if self.user_intent else None

    def update_with_evaluation(
        self, score: AggregatedScore, critiques: Optional[dict] = None
    ) -> None:
        self.stats.judge_scores = score.individual_scores
        self.stats.aggregated_score = score.aggregated_score
        if critiques:
            self.stats.critiques = critiques


# ---------------------------------------------------------------------------
# Token accounting (ref types.py:84-295, with costing removed)
# ---------------------------------------------------------------------------

TOKEN_PHASES = (
    "strategy_generation",
    "intent_generation",
    "user_simulation",
    "assistant_generation",
    "judging",
    "research",
)

#: short phase tag (as used at call sites) -> tracker attribute
PHASE_MAP = {
    "strategy": "strategy_generation",
    "intent": "intent_generation",
    "user": "user_simulation",
    "rephrase": "user_simulation",
    "assistant": "assistant_generation",
    "judge": "judging",
    "research": "research",
}


@dataclass
class TokenStats:
    input_tokens: int = 0
    output_tokens: int = 0
    total_tokens: int = 0
    request_count: int = 0

    def add(self, usage: Optional[Usage]) -> None:
        if usage:
            self.input_tokens += usage.prompt_tokens
            self.output_tokens += usage.completion_tokens
            self.total_tokens += usage.total_tokens
            self.request_count += 1


@dataclass
class TokenTracker:
    """Per-phase and per-model token books (ref types.py:118-295)."""

    model_name: str = "unknown"
    phases: dict = field(
        default_factory=lambda: {p: TokenStats() for p in TOKEN_PHASES}
    )
    by_model: dict = field(default_factory=dict)

    def add_usage(self, model: str, usage: Optional[Usage], phase: str) -> None:
        if not usage:
            return
        attr = PHASE_MAP.get(phase, phase)
        if attr in self.phases:
            self.phases[attr].add(usage)
        self.by_model.setdefault(model, TokenStats()).add(usage)

    @property
    def total_input_tokens(self) -> int:
        return sum(s.input_tokens for s in self.phases.values())

    @property
    def total_output_tokens(self) -> int:
        return sum(s.output_tokens for s in self.phases.values())

    @property
    def total_tokens(self) -> int:
        return self.total_input_tokens + self.total_output_tokens

    @property
    def total_requests(self) -> int:
        return sum(s.request_count for s in self.phases.values())

    def to_dict(self) -> dict:
        by_model = {
            name: {
                "input_tokens": s.input_tokens,
                "output_tokens": s.output_tokens,
                "requests": s.request_count,
                "cost_usd": 0.0,
            }
            for name, s in self.by_model.items()
        }
        by_phase = {
            phase: {
                "input_tokens": s.input_tokens,
                "output_tokens": s.output_tokens,
                "requests": s.request_count,
            }
            for phase, s in self.phases.items()
        }
        return {
            "models_used": list(self.by_model.keys()),
            "totals": {
                "input_tokens": self.total_input_tokens,
                "output_tokens": self.total_output_tokens,
                "total_tokens": self.total_tokens,
                "total_requests": self.total_requests,
                "total_cost_usd": 0.0,
            },
            "by_model": by_model,
            "by_phase": by_phase,
        }

    def summary_str(self) -> str:
        lines = [
            "TOKEN USAGE SUMMARY",
            f"  total in/out: {self.total_input_tokens:,} / {self.total_output_tokens:,}"
            f"  requests: {self.total_requests}",
        ]
        for phase in TOKEN_PHASES:
            s = self.phases[phase]
            if s.request_count:
                lines.append(
                    f"  {phase:<22} {s.request_count:>4} reqs  "
                    f"{s.input_tokens:>9,} in  {s.output_tokens:>9,} out"
                )
        return "\n".join(lines)


# ---------------------------------------------------------------------------
# Run result + exploration JSON checkpoint (ref types.py:439-563)
# ---------------------------------------------------------------------------


@dataclass
class DTSRunResult:
    best_node_id: Optional[str] = None
    best_score: float = 0.0
    best_messages: list = field(default_factory=list)
    all_nodes: list = field(default_factory=list)
    pruned_count: int = 0
    total_rounds: int = 0
    research_report: Optional[str] = None
    token_usage: Optional[dict] = None

    def to_exploration_dict(self) -> dict:
        """The tree-state JSON checkpoint — schema kept byte-compatible with
        ref types.py:457-554 (summary / research_report / best_branch /
        branches[...] with strategy, user_intent, status, depth, scores,
        trajectory, prune_reason). Additive extensions (ignored by the
        reference frontend, required for DTSEngine.run(resume_from=...)):
        branch-level parent_id and user_intent id/description."""
        branches = []
        for node in self.all_nodes:
            if node.strategy is None:
                continue  # skip root
            branches.append(
                {
                    "id": node.id,
                    "parent_id": node.parent_id,
                    "strategy": {
                        "tagline": node.strategy.tagline,
                        "description": node.strategy.description,
                    },
                    "user_intent": (
                        {
                            "id": node.user_intent.id,
                            "label": node.user_intent.label,
                            "description": node.user_intent.description,
                            "emotional_tone": node.user_intent.emotional_tone,
                            "cognitive_stance": node.user_intent.cognitive_stance,
                        }
                        if node.user_intent
                        else None
                    ),
                    "status": node.status.value,
                    "depth": node.depth,
                    "scores": {
                        "individual": node.stats.judge_scores,
                        "aggregated": node.stats.aggregated_score,
                        "visits": node.stats.visits,
                        "value_mean": node.stats.value_mean,
                        "critiques": node.stats.critiques or None,
                    },
                    "trajectory": [
                        {"role": m.role, "content": m.content} for m in node.messages
                    ],
                    "prune_reason": node.prune_reason,
                }
            )
        branches.sort(key=lambda b: b["scores"]["aggregated"], reverse=True)

        best_branch = None
        if self.best_node_id:
            for node in self.all_nodes:
                if node.id == self.best_node_id:
                    best_branch = {
                        "id": node.id,
                        "strategy": node.strategy.tagline if node.strategy else "root",
                        "score": self.best_score,
                        "trajectory": [
                            {"role": m.role, "content": m.content}
                            for m in node.messages
                        ],
                    }
                    break

        active = sum(1 for n in self.all_nodes if n.status == NodeStatus.ACTIVE)
        pruned = sum(1 for n in self.all_nodes if n.status == NodeStatus.PRUNED)
        result = {
            "summary": {
                "total_branches": len(branches),
                "active_branches": active,
                "pruned_branches": pruned,
                "total_rounds": self.total_rounds,
                "best_score": self.best_score,
            },
            "research_report": self.research_report,
            "best_branch": best_branch,
            "branches": branches,
        }
        if self.token_usage:
            result["token_usage"] = self.token_usage
        return result

    def to_json(self, indent: int = 2) -> str:
        return json.dumps(self.to_exploration_dict(), indent=indent, ensure_ascii=False)

    def save_json(self, path: str) -> None:
        Path(path).write_text(self.to_json(), encoding="utf-8")


def format_message_history(messages: list) -> str:
    """Flatten a trajectory for judge prompts (ref core/dts/utils.py:33-48)."""
    return "\n\n".join(
        f"{m.role.capitalize()}: {m.content or ''}" for m in messages
    )
