"""Prompt-template contract tests (parity: ref tests/core/test_prompts.py)."""

from dts_amd.search import prompts
from dts_amd.search.prompts import JUDGE_CRITERIA


class TestStructure:
    def test_all_templates_return_pairs(self):
        cases = [
            prompts.conversation_tree_generator(3, "g", "c"),
            prompts.user_intent_generator(2, "g", "h"),
            prompts.user_simulation("g"),
            prompts.assistant_continuation("g", "t", "d"),
            prompts.rephrase_with_intent("m", "l", "d", "e", "c"),
            prompts.trajectory_outcome_judge("g", "h"),
            prompts.comparative_trajectory_judge("g", []),
            prompts.branch_selection_judge("g", "c", "t", "d"),
            prompts.research_query_distill("g", "m"),
        ]
        for system, user in cases:
            assert isinstance(system, str) and isinstance(user, str)
            assert system.startswith("[dts:")


class TestContracts:
    def test_strategy_count_embedded(self):
        _, user = prompts.conversation_tree_generator(7, "goal", "ctx")
        assert "exactly 7" in user
        assert '"nodes"' in user

    def test_intent_enums_present(self):
        _, user = prompts.user_intent_generator(3, "g", "h")
        for tone in ("engaged", "skeptical", "anxious"):
            assert tone in user
        for stance in ("accepting", "challenging", "withdrawing"):
            assert stance in user

    def test_judge_rubric_has_ten_criteria(self):
        _, user = prompts.trajectory_outcome_judge("g", "history")
        assert len(JUDGE_CRITERIA) == 10
        for c in JUDGE_CRITERIA:
            assert c in user
        assert "total_score" in user

    def test_comparative_rank_schedule(self):
        system, user = prompts.comparative_trajectory_judge(
            "g",
            [{"id": "abc", "intent_label": "x", "history": "U: hi"}],
        )
        # the forced-ranking score schedule (ref prompts.py:338-344)
        assert "7.5" in system and "6.0" in system and "4.5" in system
        assert "--- Trajectory abc" in user

    def test_research_context_injection(self):
        _, without = prompts.trajectory_outcome_judge("g", "h")
        _, with_ctx = prompts.trajectory_outcome_judge("g", "h", "IMPORTANT FACTS")
        assert "IMPORTANT FACTS" not in without
        assert "IMPORTANT FACTS" in with_ctx

    def test_user_simulation_intent_conditioning(self):
        sys_plain, _ = prompts.user_simulation("g")
        sys_intent, _ = prompts.user_simulation(
            "g",
            {
                "label": "Doubter",
                "description": "d",
                "emotional_tone": "skeptical",
                "cognitive_stance": "challenging",
            },
        )
        assert "Doubter" not in sys_plain
        assert "Doubter" in sys_intent and "skeptical" in sys_intent
