"""Prompt templates for every search phase.

Parity: reference backend/core/prompts.py:10-398 — seven templates, each
returning a (system, user) pair:
  conversation_tree_generator (:22-62), user_intent_generator (:68-107),
  user_simulation (:113-149), assistant_continuation (:155-183),
  rephrase_with_intent (:189-211), trajectory_outcome_judge (:217-273),
  comparative_trajectory_judge (:329-395), branch_selection_judge (:279-323).

The wording here is our own; the *contracts* are preserved: JSON output
schemas ({"nodes": {tagline: description}}, {"intents": [...]},
{"criteria", "total_score", ...}, {"ranking", "critiques", ...}), the
10-criteria 0-1 rubric, and the comparative rank→score schedule
7.5 / 6.0 / 4.5 / −1.5-per-rank (ref prompts.py:338-344).

Every system prompt opens with a stable `[dts:<phase>]` marker: the local
engine uses it for phase-aware scheduling hints and the deterministic fake
backend keys on it (dts_amd/llm/fake.py).
"""

from __future__ import annotations

from typing import Optional


def conversation_tree_generator(
    num_nodes: int,
    conversation_goal: str,
    conversation_context: str,
    deep_research_context: Optional[str] = None,
) -> tuple:
    system = (
        "[dts:strategy] You design opening strategies for goal-directed "
        "conversations. Produce genuinely distinct approaches — different "
        "framings, orderings and emotional registers, not paraphrases of one "
        "idea. Respond with a single valid JSON object and nothing else: no "
        "markdown fences, no commentary."
    )
    research = (
        f"\n\nBackground research to draw on:\n{deep_research_context}\n"
        if deep_research_context
        else ""
    )
    user = (
        f"Goal of the conversation: {conversation_goal}\n\n"
        f"The user opens with: {conversation_context}\n{research}\n"
        f"Propose exactly {num_nodes} distinct conversation strategies.\n\n"
        "Return JSON of this shape:\n"
        "{\n"
        '  "goal": "restated goal",\n'
        '  "nodes": {"<short tagline>": "<2-3 sentence description of the strategy>", ...},\n'
        '  "coverage_rationale": "why these strategies span the space"\n'
        "}\n"
        f'The "nodes" object must contain exactly {num_nodes} entries.'
    )
    return system, user


def conversation_tree_generator_single(
    index: int,
    total: int,
    conversation_goal: str,
    conversation_context: str,
    lens: str,
    deep_research_context: Optional[str] = None,
) -> tuple:
    """Split strategy generation: one strategy per call, all calls
    sharing the (system + goal + context) prompt prefix; a per-call
    diversity lens replaces the combined form's see-your-siblings
    diversity pressure. Sequential decode depth drops from one
    N-strategy form to one strategy."""
    system = (
        "[dts:strategy] You design opening strategies for goal-directed "
        "conversations. Produce genuinely distinct approaches — different "
        "framings, orderings and emotional registers, not paraphrases of one "
        "idea. Respond with a single valid JSON object and nothing else: no "
        "markdown fences, no commentary."
    )
    research = (
        f"\n\nBackground research to draw on:\n{deep_research_context}\n"
        if deep_research_context
        else ""
    )
    user = (
        f"Goal of the conversation: {conversation_goal}\n\n"
        f"The user opens with: {conversation_context}\n{research}\n"
        f"This is strategy {index} of {total}. Approach it strictly "
        f"through the lens of: {lens}. Other calls cover other lenses.\n\n"
        "Return JSON of this shape:\n"
        "{\n"
        '  "goal": "restated goal",\n'
        '  "nodes": {"<short tagline>": "<2-3 sentence description of the strategy>"},\n'
        '  "coverage_rationale": "why this angle matters"\n'
        "}\n"
        'The "nodes" object must contain exactly 1 entries.'
    )
    return system, user


def user_intent_generator(
    num_intents: int,
    conversation_goal: str,
    conversation_history: str,
) -> tuple:
    system = (
        "[dts:intent] You model how different real users might plausibly "
        "respond next in a conversation. Vary emotional tone and cognitive "
        "stance; avoid near-duplicates. Respond with a single valid JSON "
        "object only — no fences, no prose."
    )
    user = (
        f"Goal: {conversation_goal}\n\n"
        f"Conversation so far:\n{conversation_history}\n\n"
        f"Produce exactly {num_intents} distinct user intents for the next "
        "user turn.\n\n"
        "Return JSON of this shape:\n"
        "{\n"
        '  "intents": [\n'
        "    {\n"
        '      "id": "snake_case_id",\n'
        '      "label": "Short Label",\n'
        '      "description": "one sentence",\n'
        '      "emotional_tone": "engaged|resistant|confused|skeptical|enthusiastic|deflecting|anxious|neutral",\n'
        '      "cognitive_stance": "accepting|questioning|challenging|exploring|withdrawing"\n'
        "    }, ...\n"
        "  ]\n"
        "}"
    )
    return system, user


def user_simulation(
    conversation_goal: str,
    user_intent: Optional[dict] = None,
) -> tuple:
    intent_block = ""
    if user_intent:
        intent_block = (
            "\nAdopt this persona for your reply:\n"
            f"- intent: {user_intent.get('label')} — {user_intent.get('description')}\n"
            f"- emotional tone: {user_intent.get('emotional_tone')}\n"
            f"- cognitive stance: {user_intent.get('cognitive_stance')}\n"
        )
    system = (
        "[dts:user-sim] You play the USER in an ongoing conversation. Stay "
        "in character, react naturally to the assistant's last message, and "
        "keep replies conversational (a few sentences). Never break the "
        f"fourth wall. The conversation's underlying topic: {conversation_goal}."
        f"{intent_block}"
    )
    user = "Write the user's next message, and only that message."
    return system, user


def assistant_continuation(
    conversation_goal: str,
    strategy_tagline: str,
    strategy_description: str,
) -> tuple:
    system = (
        "[dts:assistant] You are the ASSISTANT in an ongoing conversation. "
        f"Your objective: {conversation_goal}.\n"
        f"Follow this strategy — {strategy_tagline}: {strategy_description}\n"
        "Be concrete, move the conversation forward every turn, and keep "
        "replies focused."
    )
    user = "Write the assistant's next message, and only that message."
    return system, user


def rephrase_with_intent(
    original_message: str,
    intent_label: str,
    intent_description: str,
    emotional_tone: str,
    cognitive_stance: str,
) -> tuple:
    system = (
        "[dts:rephrase] Rewrite a user's opening message so it expresses a "
        "given persona while preserving the underlying request. Output only "
        "the rewritten message."
    )
    user = (
        f"Original message: {original_message}\n\n"
        f"Persona: {intent_label} — {intent_description}\n"
        f"Emotional tone: {emotional_tone}\n"
        f"Cognitive stance: {cognitive_stance}\n\n"
        "Rewritten message:"
    )
    return system, user


#: the 10 criteria of the absolute judge rubric (ref prompts.py:246-257)
JUDGE_CRITERIA = (
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
)


def trajectory_outcome_judge(
    conversation_goal: str,
    conversation_history: str,
    deep_research_context: Optional[str] = None,
) -> tuple:
    system = (
        "[dts:judge-absolute] You are a strict evaluator of complete "
        "conversation trajectories. Hunt for flaws and missed opportunities; "
        "most conversations deserve middling scores. Calibration: 7/10 is "
        "genuinely good, 8+ is rare, 10 is almost never warranted. Respond "
        "with one valid JSON object only — no fences, no prose."
    )
    research = (
        f"\nResearch context (judge whether choices were well-informed):\n"
        f"{deep_research_context}\n"
        if deep_research_context
        else ""
    )
    criteria_lines = "\n".join(
        f"{i + 1}. {name}" for i, name in enumerate(JUDGE_CRITERIA)
    )
    user = (
        f"Goal: {conversation_goal}\n\n"
        f"Conversation:\n{conversation_history}\n{research}\n"
        "Score each criterion from 0.0 to 1.0 (find something to critique in "
        "each):\n"
        f"{criteria_lines}\n\n"
        "Return JSON of this shape:\n"
        "{\n"
        '  "criteria": {"goal_achieved": {"score": 0.0, "rationale": "..."}, ...},\n'
        '  "total_score": <sum of criterion scores, 0-10>,\n'
        '  "confidence": "low|medium|high",\n'
        '  "summary": "one-sentence critique",\n'
        '  "key_turning_point": "the decisive moment",\n'
        '  "biggest_missed_opportunity": "what would have improved it"\n'
        "}\n"
        "Sanity-check: totals typically land between 4 and 7."
    )
    return system, user


_COMPARATIVE_SYSTEM = (
    "[dts:judge-comparative] You force-rank sibling conversation "
    "trajectories against each other. No ties. Use this score schedule: "
    "rank 1 → 7.5, rank 2 → 6.0, rank 3 → 4.5, each further rank 1.5 "
    "lower. Only raise rank 1 above 8.0 for truly exceptional execution. "
    "Respond with one valid JSON object only."
)


def _comparative_user_base(
    conversation_goal: str,
    trajectories: list,
    deep_research_context: Optional[str],
) -> str:
    """Shared (system + goal + all trajectories) prefix for the combined
    AND the split comparative calls — every split call re-renders this
    exact prefix so the paged KV prefix cache serves its prefill."""
    research = (
        f"\nResearch context:\n{deep_research_context}\n"
        if deep_research_context
        else ""
    )
    blocks = []
    for t in trajectories:
        blocks.append(
            f"--- Trajectory {t['id']} (intent: {t.get('intent_label', 'unknown')}) ---\n"
            f"{t['history']}"
        )
    traj_text = "\n\n".join(blocks)
    return f"Goal: {conversation_goal}\n{research}\nTrajectories:\n{traj_text}\n\n"


def comparative_trajectory_judge(
    conversation_goal: str,
    trajectories: list,
    deep_research_context: Optional[str] = None,
) -> tuple:
    user = _comparative_user_base(
        conversation_goal, trajectories, deep_research_context
    ) + (
        "For each trajectory list 2-3 concrete weaknesses and at least one "
        "strength, then force-rank all of them.\n\n"
        "Return JSON of this shape:\n"
        "{\n"
        '  "critiques": {"<trajectory_id>": {"weaknesses": [...], "strengths": [...], "key_moment": "..."}},\n'
        '  "ranking": [{"rank": 1, "trajectory_id": "...", "score": 7.5, "reason": "..."}, ...],\n'
        '  "ranking_confidence": "low|medium|high"\n'
        "}"
    )
    return _COMPARATIVE_SYSTEM, user


def comparative_critique_judge(
    conversation_goal: str,
    trajectories: list,
    target_id: str,
    deep_research_context: Optional[str] = None,
) -> tuple:
    """Split comparative mode, critique leg: same full sibling context
    (so the critique stays comparative AND the prompt prefix is shared),
    critiquing exactly one trajectory."""
    user = _comparative_user_base(
        conversation_goal, trajectories, deep_research_context
    ) + (
        f"Critique ONLY trajectory {target_id}: list 2 concrete "
        "weaknesses, at least one strength, and the key turning moment.\n"
        "Return JSON of this shape:\n"
        '{"weaknesses": ["...", "..."], "strengths": ["..."], '
        '"key_moment": "..."}\n'
        "[dts:part=critique]"
    )
    return _COMPARATIVE_SYSTEM, user


def comparative_ranking_judge(
    conversation_goal: str,
    trajectories: list,
    deep_research_context: Optional[str] = None,
) -> tuple:
    """Split comparative mode, ranking leg (critiques travel separately)."""
    user = _comparative_user_base(
        conversation_goal, trajectories, deep_research_context
    ) + (
        "Force-rank ALL trajectories (critiques are collected "
        "separately; rank and score only).\n"
        "Return JSON of this shape:\n"
        '{"ranking": [{"rank": 1, "trajectory_id": "...", "score": 7.5, '
        '"reason": "..."}, ...], "ranking_confidence": "low|medium|high"}\n'
        "[dts:part=ranking]"
    )
    return _COMPARATIVE_SYSTEM, user


def branch_selection_judge(
    conversation_goal: str,
    conversation_context: str,
    branch_tagline: str,
    branch_description: str,
) -> tuple:
    """Pre-exploration branch scoring. Defined for parity with ref
    prompts.py:279-323; the reference engine never calls it (SURVEY.md
    §4.1.5) and neither does ours."""
    system = (
        "[dts:judge-branch] You score how promising a conversation direction "
        "is before it is explored — position, not outcome. Respond with one "
        "valid JSON object only."
    )
    user = (
        f"Goal: {conversation_goal}\n\nContext:\n{conversation_context}\n\n"
        f"Proposed branch: {branch_tagline} — {branch_description}\n\n"
        "Score these ten criteria with 0, 0.5 or 1 each: goal_aligned, "
        "contextually_appropriate, emotionally_attuned, well_timed, "
        "builds_on_history, information_generating, not_redundant, "
        "appropriately_scoped, actionable, low_risk.\n\n"
        'Return JSON: {"criteria": {...}, "total_score": 0-10, '
        '"confidence": "low|medium|high", "summary": "..."}'
    )
    return system, user


def research_query_distill(goal: str, first_message: str) -> tuple:
    """Distill a research query (ref researcher.py:36-41, 241-261)."""
    system = (
        "[dts:research-query] Turn a conversation goal into one focused web "
        "research query. Output only the query text."
    )
    user = f"Goal: {goal}\nOpening message: {first_message}\n\nResearch query:"
    return system, user
