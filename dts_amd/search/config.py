"""Search configuration.

Parity: reference backend/core/dts/config.py:21-69 (full knob set, same
defaults). Additions for the local MI355X engine: per-phase generation
budgets (the reference sets no max_tokens — SURVEY.md §4.1.7 — which an
in-process scheduler cannot tolerate) and a seed for deterministic runs.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional


@dataclass
class GenerationBudget:
    """Per-phase max new tokens for the local engine."""

    strategy: int = 1024
    intent: int = 768
    rephrase: int = 160
    user: int = 160
    assistant: int = 256
    judge: int = 1024


@dataclass
class DTSConfig:
    goal: str
    first_message: str
    init_branches: int = 6
    deep_research: bool = False
    research_cache_dir: str = ".cache/research"
    # mid-run crash recovery (framework addition — the reference persists
    # the tree-state JSON only at the end): after every round the engine
    # atomically writes the exploration dict here; resume with
    # DTSEngine.run(resume_from=checkpoint_path). DP: rank 0 writes.
    checkpoint_path: Optional[str] = None
    turns_per_branch: int = 5
    user_intents_per_branch: int = 3
    user_variability: bool = False
    scoring_mode: str = "comparative"  # "absolute" | "comparative"
    prune_threshold: float = 6.5
    keep_top_k: Optional[int] = None
    min_survivors: int = 1
    max_concurrency: int = 16
    model: Optional[str] = None
    strategy_model: Optional[str] = None
    simulator_model: Optional[str] = None
    judge_model: Optional[str] = None
    temperature: float = 0.7
    judge_temperature: float = 0.3
    reasoning_enabled: bool = False
    provider: Optional[str] = None
    # --- local-engine additions ---
    budget: GenerationBudget = field(default_factory=GenerationBudget)
    seed: Optional[int] = None
    # split comparative judging: parallel per-sibling critique calls +
    # one ranking-only call, all sharing one cached prompt prefix — cuts
    # the score phase's sequential decode depth ~4x on the local engine.
    # False reproduces the reference's single combined ranking call.
    comparative_split: bool = True
    # split strategy generation: one parallel call per strategy through
    # a distinct diversity lens (shared prompt prefix) instead of the
    # reference's single N-node form — cuts the init phase's sequential
    # decode depth ~N x on the local engine.
    strategy_split: bool = True

    def __post_init__(self) -> None:
        if self.scoring_mode not in ("absolute", "comparative"):
            raise ValueError(f"invalid scoring_mode: {self.scoring_mode}")
