"""Median-vote tests (parity with ref tests/core/dts/test_aggregator.py:78-107)."""

import pytest

from dts_amd.search import aggregate_majority_vote


def test_median_of_three():
    agg = aggregate_majority_vote([3.0, 9.0, 6.0], pass_threshold=5.0)
    assert agg.aggregated_score == 6.0
    assert agg.individual_scores == [3.0, 9.0, 6.0]


def test_pass_requires_two_votes():
    agg = aggregate_majority_vote([6.0, 6.0, 2.0], pass_threshold=5.0)
    assert agg.pass_votes == 2
    assert agg.passed

    agg = aggregate_majority_vote([6.0, 2.0, 2.0], pass_threshold=5.0)
    assert agg.pass_votes == 1
    assert not agg.passed


def test_exact_threshold_counts_as_pass():
    agg = aggregate_majority_vote([5.0, 5.0, 1.0], pass_threshold=5.0)
    assert agg.pass_votes == 2
    assert agg.passed


def test_wrong_arity_raises():
    with pytest.raises(ValueError):
        aggregate_majority_vote([1.0, 2.0])
    with pytest.raises(ValueError):
        aggregate_majority_vote([1.0, 2.0, 3.0, 4.0])


def test_zero_score_shape():
    from dts_amd.search import AggregatedScore

    z = AggregatedScore.zero(6.5)
    assert z.individual_scores == [0.0, 0.0, 0.0]
    assert z.pass_threshold == 6.5
    assert not z.passed
