"""Wire schemas for the WS/REST server.

Parity: reference backend/api/schemas.py:12-107. `SearchRequest` is the
compatibility contract (same fields, same bounds). Additively extended
with `user_variability` and `reasoning_enabled` — the reference frontend
already sends both but its backend silently dropped them (SURVEY.md
§4.1.1); declaring them here closes that gap while staying
wire-compatible.
"""

from __future__ import annotations

from typing import Literal, Optional

from pydantic import BaseModel, Field

ScoringMode = Literal["absolute", "comparative"]


class SearchRequest(BaseModel):
    goal: str = Field(..., description="Conversation goal/objective")
    first_message: str = Field(..., description="Initial user message")
    init_branches: int = Field(default=6, ge=1, le=20)
    turns_per_branch: int = Field(default=5, ge=1, le=20)
    user_intents_per_branch: int = Field(default=3, ge=1, le=10)
    scoring_mode: ScoringMode = Field(default="comparative")
    prune_threshold: float = Field(default=6.5, ge=0.0, le=10.0)
    rounds: int = Field(default=1, ge=1, le=10)
    deep_research: bool = Field(default=False)
    strategy_model: Optional[str] = Field(default=None)
    simulator_model: Optional[str] = Field(default=None)
    judge_model: Optional[str] = Field(default=None)
    # additive fields (closed schema gap)
    user_variability: bool = Field(default=False)
    reasoning_enabled: bool = Field(default=False)
    # additive: a saved exploration dict to resume from (run continues
    # `rounds` more rounds on the restored tree instead of generating
    # fresh strategies)
    resume_from: Optional[dict] = Field(default=None)
    # additive: per-round crash-recovery checkpoint file (see
    # DTSConfig.checkpoint_path)
    checkpoint_path: Optional[str] = Field(default=None)


class EventMessage(BaseModel):
    type: str
    data: dict = Field(default_factory=dict)


class ErrorData(BaseModel):
    message: str
    code: Optional[str] = None


class SearchStartedData(BaseModel):
    goal: str
    first_message: str
    total_rounds: int
    config: dict


class PhaseData(BaseModel):
    phase: str
    message: str


class StrategyGeneratedData(BaseModel):
    index: int
    total: int
    tagline: str
    description: str


class NodeAddedData(BaseModel):
    id: str
    parent_id: Optional[str]
    depth: int
    status: str
    strategy: Optional[str]
    user_intent: Optional[str]
    message_count: int


class NodeUpdatedData(BaseModel):
    id: str
    status: str
    score: float
    individual_scores: list
    passed: bool


class RoundStartedData(BaseModel):
    round: int
    total_rounds: int


class NodesPrunedData(BaseModel):
    ids: list
    reasons: dict


class IntentGeneratedData(BaseModel):
    strategy: str
    index: int
    total: int
    label: str
    emotional_tone: str
    cognitive_stance: str


class TokenUpdateData(BaseModel):
    totals: dict


class CompleteData(BaseModel):
    best_node_id: Optional[str]
    best_score: float
    pruned_count: int
    total_rounds: int
    exploration: dict
