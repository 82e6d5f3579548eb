"""FormGuide FSM unit tests (no engine): segment walking, budget caps,
choice tries, no-repeat ranking ids."""

import json

import pytest

from dts_amd.serving.structured import (
    Choice,
    Fixed,
    FormGuide,
    Free,
    comparative_judge_form,
    intent_form,
    strategy_form,
)
from dts_amd.serving.tokenizer import SyntheticTokenizer


@pytest.fixture
def tok():
    return SyntheticTokenizer(1024)


def drive(tok, guide, pick=None):
    """Run a guide to completion, choosing the FIRST allowed token each
    step (or via `pick`); returns the produced text."""
    out = list(guide.initial_forced())
    steps = 0
    while not guide.done():
        allowed = guide.allowed_tokens()
        assert allowed, "guide must constrain every sampled step"
        t = pick(allowed, steps) if pick else allowed[0]
        out.append(t)
        out.extend(guide.on_token(t))
        steps += 1
        assert steps < 20000
    return tok.decode(out)


class TestSegments:
    def test_fixed_free_fixed(self, tok):
        g = FormGuide(tok, [Fixed('{"a": "'), Free(max_tokens=4, stop=ord('"')),
                            Fixed("}")])
        text = drive(tok, g, pick=lambda a, s: ord("x") if s < 2 else ord('"'))
        assert json.loads(text.replace("}", "}"))  # valid JSON
        assert text == '{"a": "xx"}'

    def test_budget_forces_terminator(self, tok):
        g = FormGuide(tok, [Fixed('{"a": "'), Free(max_tokens=3, stop=ord('"')),
                            Fixed("}")])
        # never sample the quote: budget exhausts, guide closes the string
        text = drive(tok, g, pick=lambda a, s: ord("y"))
        assert text == '{"a": "yyy"}'
        assert json.loads(text)

    def test_choice_walks_trie(self, tok):
        g = FormGuide(tok, [Fixed('"'), Choice(["low", "medium", "high"]), Fixed('"')])

        def pick(allowed, step):
            # steer toward "medium"
            want = b"medium"
            for b in allowed:
                if step < len(want) and b == want[step]:
                    return b
            return allowed[0]

        text = drive(tok, g, pick=pick)
        assert text == '"medium"'


class TestForms:
    def test_strategy_form_unique_keys(self, tok):
        g = strategy_form(tok, 5)
        text = drive(tok, g, pick=lambda a, s: a[-1])
        obj = json.loads(text)
        assert len(obj["nodes"]) == 5  # fixed unique key prefixes guarantee it

    def test_intent_form_enums(self, tok):
        g = intent_form(tok, 3)
        text = drive(tok, g)
        obj = json.loads(text)
        assert len(obj["intents"]) == 3
        for it in obj["intents"]:
            assert it["emotional_tone"] in (
                "engaged", "resistant", "confused", "skeptical",
                "enthusiastic", "deflecting", "anxious", "neutral",
            )

    def test_ranking_no_repeats_any_sampling(self, tok):
        ids = [f"{c}0000000-0000-0000-0000-00000000000{i}"
               for i, c in enumerate("abcd")]
        # adversarial: always pick the LAST allowed byte
        g = comparative_judge_form(tok, ids)
        text = drive(tok, g, pick=lambda a, s: a[-1])
        obj = json.loads(text)
        ranked = [r["trajectory_id"] for r in obj["ranking"]]
        assert sorted(ranked) == sorted(ids)
        assert [r["rank"] for r in obj["ranking"]] == [1, 2, 3, 4]


class TestRandomDrivesProduceValidJson:
    """Property: ANY token stream the guide permits decodes to JSON that
    parses and matches the phase schema — the guarantee the search layer
    relies on when sampling from random-init weights."""

    @pytest.mark.parametrize("seed", range(5))
    def test_strategy_form(self, tok, seed):
        import random

        rng = random.Random(seed)
        text = drive(tok, strategy_form(tok, 4), pick=lambda a, s: rng.choice(a))
        d = json.loads(text)
        assert set(d) == {"goal", "nodes", "coverage_rationale"}
        assert len(d["nodes"]) == 4
        for i, key in enumerate(d["nodes"], start=1):
            assert key.startswith(f"Strategy {i}: ")

    @pytest.mark.parametrize("seed", range(5))
    def test_intent_form(self, tok, seed):
        import random

        rng = random.Random(seed)
        text = drive(tok, intent_form(tok, 3), pick=lambda a, s: rng.choice(a))
        d = json.loads(text)
        assert len(d["intents"]) == 3
        for n, i in enumerate(d["intents"], start=1):
            assert i["id"] == f"intent_{n}"
            assert set(i) == {
                "id", "label", "description", "emotional_tone", "cognitive_stance",
            }

    @pytest.mark.parametrize("seed", range(5))
    def test_absolute_judge_form(self, tok, seed):
        import random

        from dts_amd.serving.structured import absolute_judge_form

        rng = random.Random(seed)
        text = drive(tok, absolute_judge_form(tok), pick=lambda a, s: rng.choice(a))
        d = json.loads(text)
        assert len(d["criteria"]) == 10
        for c in d["criteria"].values():
            assert 0.0 <= float(c["score"]) <= 1.0
        assert "total_score" in d

    @pytest.mark.parametrize("seed", range(8))
    def test_ranking_guide_permutation(self, tok, seed):
        import random

        ids = [f"00000000-0000-0000-0000-{i:012d}" for i in range(5)]
        rng = random.Random(seed)
        text = drive(
            tok, comparative_judge_form(tok, ids), pick=lambda a, s: rng.choice(a)
        )
        d = json.loads(text)
        ranked = [r["trajectory_id"] for r in d["ranking"]]
        assert sorted(ranked) == sorted(ids)  # a true permutation, no repeats
