"""Simulator tests (parity coverage of ref tests/.../test_simulator.py)."""

import json

import pytest

from dts_amd.llm import LLM, FakeBackend, Message, ScriptedBackend
from dts_amd.search.generator import FIXED_INTENT
from dts_amd.search.simulator import ConversationSimulator, TERMINATION_SIGNALS
from dts_amd.search.tree import DialogueTree, generate_node_id
from dts_amd.search.types import DialogueNode, NodeStatus, Strategy, UserIntent


def make_sim(backend, **kw):
    return ConversationSimulator(LLM(backend, default_model="m"), goal="g", **kw)


def make_node(msg="hello, can you help me learn this?"):
    return DialogueNode(
        id=generate_node_id(),
        strategy=Strategy(tagline="t", description="d"),
        messages=[Message.user(msg)],
    )


INTENT = UserIntent(
    id="i1", label="L", description="D", emotional_tone="engaged", cognitive_stance="exploring"
)


class TestTermination:
    @pytest.mark.parametrize("signal", ["goodbye", "stop", "i give up", "never mind"])
    def test_signals_terminate(self, signal):
        sim = make_sim(FakeBackend())
        assert sim._should_terminate(f"ok then, {signal}!")

    def test_short_frustrated(self):
        sim = make_sim(FakeBackend())
        assert sim._should_terminate("nope.")
        assert not sim._should_terminate("This is a wonderful detailed answer thanks a lot")

    def test_terminating_user_marks_terminal(self, run_async):
        # user reply contains a termination signal -> node TERMINAL, no assistant reply
        backend = ScriptedBackend(["thanks, bye"])
        sim = make_sim(backend)
        node = make_node()
        result = run_async(sim._expand_linear(node, turns=3))
        assert result.status == NodeStatus.TERMINAL
        # one user message appended, no assistant
        assert [m.role for m in result.messages] == ["user", "user"]


class TestLinearExpansion:
    def test_turn_structure(self, run_async):
        sim = make_sim(FakeBackend())
        node = make_node()
        result = run_async(sim._expand_linear(node, turns=2))
        roles = [m.role for m in result.messages]
        assert roles == ["user", "user", "assistant", "user", "assistant"]
        assert result.status == NodeStatus.ACTIVE

    def test_batch_collects_errors(self, run_async):
        # all three calls for node 1's first user sim are empty -> error status
        backend = ScriptedBackend(["", "", ""])
        sim = make_sim(backend)
        node = make_node()
        expanded = run_async(sim._expand_linear_batch([node], turns=1))
        assert expanded == [node]  # node returned with error status
        assert node.status == NodeStatus.ERROR
        assert "empty user response" in node.prune_reason


class TestEmptyRetry:
    def test_retries_then_succeeds(self, run_async):
        backend = ScriptedBackend(["", "  ", "real answer"])
        sim = make_sim(backend)
        out = run_async(
            sim._call_with_retry([Message.user("x")], phase="user", max_tokens=32)
        )
        assert out == "real answer"
        assert len(backend.calls) == 3


class TestForking:
    def test_fork_creates_children_in_tree(self, run_async):
        sim = make_sim(FakeBackend())
        root = DialogueNode(id=generate_node_id(), messages=[Message.user("start")])
        tree = DialogueTree.create(root)
        parent = make_node()
        tree.add_child(root.id, parent)

        async def gen_intents(history, count):
            return [INTENT, FIXED_INTENT][:count]

        expanded = run_async(
            sim.expand_nodes(
                [parent], turns=1, intents_per_node=2, tree=tree, generate_intents=gen_intents
            )
        )
        assert len(expanded) == 2
        assert all(n.parent_id == parent.id for n in expanded)
        assert all(n.user_intent is not None for n in expanded)
        assert len(parent.children) == 2
        # rephrased first message + assistant reply on turn 0 (user sim skipped)
        for n in expanded:
            assert n.messages[0].role == "user"
            assert n.messages[1].role == "assistant"

    def test_intent_failure_falls_back_linear(self, run_async):
        sim = make_sim(FakeBackend())
        parent = make_node()

        async def failing_intents(history, count):
            raise RuntimeError("no intents")

        expanded = run_async(
            sim.expand_nodes(
                [parent], turns=1, intents_per_node=2, tree=None, generate_intents=failing_intents
            )
        )
        # falls back to linear expansion of the parent itself
        assert expanded == [parent]
        assert parent.user_intent is None

    def test_intents_per_node_one_short_circuits(self, run_async):
        """intents_per_node<=1 -> linear, generate_intents never called
        (ref simulator.py:123-125, the FIXED_INTENT-dead-code fact
        SURVEY.md §4.1.2)."""
        sim = make_sim(FakeBackend())
        parent = make_node()
        called = []

        async def gen(history, count):
            called.append(1)
            return [FIXED_INTENT]

        expanded = run_async(
            sim.expand_nodes([parent], turns=1, intents_per_node=1, generate_intents=gen)
        )
        assert expanded == [parent]
        assert not called

    def test_rephrase_failure_keeps_original(self, run_async):
        # rephrase returns empty 3x -> keep original first message, then 1 turn
        responses = ["", "", ""] + ["assistant reply"]
        backend = ScriptedBackend(responses)
        sim = make_sim(backend)
        node = make_node("original opening")
        result = run_async(sim._expand_with_intent(node, turns=1, first_intent=INTENT))
        assert result.messages[0].content == "original opening"
        assert result.messages[1].content == "assistant reply"


class TestReasoningEnabled:
    """reasoning_enabled (the wire flag the reference silently drops,
    SURVEY.md §4.1.1): assistant turns get a think-span instruction and
    <think> spans are stripped from the stored trajectory."""

    def test_think_instruction_and_strip(self):
        import asyncio

        from dts_amd.llm import LLM
        from dts_amd.llm.fake import FakeBackend
        from dts_amd.search.simulator import ConversationSimulator
        from dts_amd.search.types import DialogueNode, Strategy

        backend = FakeBackend()
        orig = backend._respond

        def with_think(system, user, messages):
            text = orig(system, user, messages)
            if "[dts:assistant]" in system:
                return f"<think>hidden chain of thought</think>{text}"
            return text

        backend._respond = with_think
        llm = LLM(backend, default_model="fake")
        sim = ConversationSimulator(
            llm, goal="g", reasoning_enabled=True, seed=1
        )
        node = DialogueNode(
            id="n1",
            strategy=Strategy(tagline="t", description="d"),
            messages=[],
        )
        from dts_amd.llm.types import Message

        history = [Message.user("hi there can you help me?")]
        ok = asyncio.run(sim._run_turn(node, history, 0, skip_user_simulation=True))
        assert ok
        assistant = history[-1]
        assert assistant.role == "assistant"
        assert "<think>" not in (assistant.content or "")
        assert "hidden chain of thought" not in (assistant.content or "")
        # the instruction reached the model
        sys_prompts = [c["system"] for c in backend.calls if "[dts:assistant]" in c["system"]]
        assert sys_prompts and "<think>" in sys_prompts[-1]


class TestTaskBudget:
    """Per-task wall budget (ref simulator.py:199-214): a hung expansion
    times out without sinking the batch."""

    def test_hung_expansion_times_out(self, run_async, monkeypatch):
        import asyncio

        import dts_amd.search.simulator as sim_mod

        monkeypatch.setattr(sim_mod, "TASK_TIMEOUT_S", 0.2)

        class _HangBackend(FakeBackend):
            def __init__(self):
                super().__init__()
                self.n = 0

            async def chat(self, messages, params, model=None):
                self.n += 1
                if self.n == 1:  # first fork's rephrase call hangs
                    await asyncio.sleep(30)
                return await super().chat(messages, params, model)

        sim = make_sim(_HangBackend())
        root = DialogueNode(id=generate_node_id(), messages=[Message.user("s")])
        tree = DialogueTree.create(root)
        a, b = make_node("first branch"), make_node("second branch")
        tree.add_child(root.id, a)
        tree.add_child(root.id, b)

        intent2 = UserIntent(
            id="i2",
            label="L2",
            description="D2",
            emotional_tone="curious",
            cognitive_stance="probing",
        )

        async def gen_intents(history, count):
            return [INTENT, intent2][:count]

        async def run():
            return await asyncio.wait_for(
                sim.expand_nodes(
                    [a, b],
                    turns=1,
                    intents_per_node=2,
                    tree=tree,
                    generate_intents=gen_intents,
                ),
                timeout=20,
            )

        expanded = run_async(run())
        # 4 fork tasks; the hung one dropped, the healthy ones completed
        assert len(expanded) == 3


class TestTerminalExclusion:
    """Terminal/pruned nodes never re-expand (ref tree.py:85-89
    active_leaves filters by NodeStatus.ACTIVE)."""

    def test_active_leaves_excludes_terminal_and_pruned(self):
        root = DialogueNode(id=generate_node_id(), messages=[Message.user("s")])
        tree = DialogueTree.create(root)
        kids = [make_node(f"k{i}") for i in range(3)]
        for k in kids:
            tree.add_child(root.id, k)
        kids[0].status = NodeStatus.TERMINAL
        kids[1].status = NodeStatus.PRUNED
        leaves = tree.active_leaves()
        assert leaves == [kids[2]]
