"""Median-vote aggregation over 3 judges.

Parity: reference backend/core/dts/aggregator.py:15-50 — median of exactly
3 scores; passed requires >=2 votes at threshold; raises on arity != 3.
"""

from __future__ import annotations

from dts_amd.search.types import AggregatedScore


def aggregate_majority_vote(
    scores: list, pass_threshold: float = 5.0
) -> AggregatedScore:
    if len(scores) != 3:
        raise ValueError(f"Expected exactly 3 scores, got {len(scores)}")
    aggregated = sorted(scores)[1]
    pass_votes = sum(1 for s in scores if s >= pass_threshold)
    return AggregatedScore(
        individual_scores=list(scores),
        aggregated_score=aggregated,
        pass_threshold=pass_threshold,
        pass_votes=pass_votes,
        passed=pass_votes >= 2,
    )
