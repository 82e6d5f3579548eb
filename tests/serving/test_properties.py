"""Property-based tests (hypothesis) for the invariant-heavy primitives.

The reference's test style is example-based mocks; these
add machine-checked invariants for the pieces that have none upstream:
the refcounted prefix-cache block manager, the n-gram draft proposer and
the constrained-decoding form guides.
"""

import json

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st
from hypothesis.stateful import RuleBasedStateMachine, invariant, precondition, rule

from dts_amd.serving.kv_cache import BlockManager, chain_hash
from dts_amd.serving.spec import NgramIndex

SETTINGS = settings(
    max_examples=60,
    deadline=None,
    derandomize=True,
    suppress_health_check=[HealthCheck.too_slow],
)


class BlockManagerMachine(RuleBasedStateMachine):
    """Model-based check: refcount conservation, eviction-pool accounting
    and prefix-match correctness under arbitrary interleavings."""

    def __init__(self):
        super().__init__()
        self.bm = BlockManager(24, block_size=4)
        self.held = []  # block ids we hold one ref to (with multiplicity)
        self.chains = []  # registered chains: list of (tokens, [block_ids])

    @precondition(lambda self: self.bm.num_free() > 0)
    @rule(data=st.data())
    def alloc_and_maybe_register(self, data):
        bid = self.bm.allocate_fresh()
        self.held.append(bid)
        if data.draw(st.booleans()):
            # grow a chain: register this block as full content
            grow = self.chains and data.draw(st.booleans())
            if grow:
                tokens, ids = data.draw(st.sampled_from(self.chains))
            else:
                tokens, ids = [], []
            chunk = tuple(
                data.draw(st.integers(min_value=0, max_value=30))
                for _ in range(self.bm.block_size)
            )
            prev = 0
            for s in range(0, len(tokens), 4):
                prev = chain_hash(prev, tuple(tokens[s : s + 4]))
            self.bm.register_full_block(bid, prev, chunk)
            self.chains.append((list(tokens) + list(chunk), ids + [bid]))

    @precondition(lambda self: self.held)
    @rule(data=st.data())
    def free_one(self, data):
        i = data.draw(st.integers(min_value=0, max_value=len(self.held) - 1))
        bid = self.held.pop(i)
        self.bm.free_block(bid)

    @precondition(lambda self: self.chains)
    @rule(data=st.data(), extra=st.lists(st.integers(0, 30), max_size=3))
    def match_registered_chain(self, data, extra):
        tokens, _ids = data.draw(st.sampled_from(self.chains))
        ids, n = self.bm.match_prefix(list(tokens) + extra)
        # every matched block's registered content is the true prefix
        assert n == len(ids) * self.bm.block_size
        assert n <= len(tokens) + len(extra)
        flat = []
        for bid in ids:
            flat.extend(self.bm.blocks[bid].token_ids)
        assert flat == list(tokens[:n]) + list(extra[: n - len(tokens)])
        self.held.extend(ids)  # matched blocks were acquired

    @invariant()
    def conservation(self):
        bm = self.bm
        refs = sum(b.ref_count for b in bm.blocks)
        assert refs == len(self.held)
        n_zero = sum(1 for b in bm.blocks if b.ref_count == 0)
        assert len(bm.free_ids) + len(bm.evictable) == n_zero
        # evictable blocks are exactly the zero-ref hashed ones
        for bid in bm.evictable:
            assert bm.blocks[bid].ref_count == 0


BlockManagerMachine.TestCase.settings = settings(
    max_examples=40, stateful_step_count=40, deadline=None, derandomize=True
)
TestBlockManagerMachine = BlockManagerMachine.TestCase


class TestNgramProperties:
    @SETTINGS
    @given(
        tokens=st.lists(st.integers(0, 8), min_size=3, max_size=120),
        max_k=st.integers(1, 8),
    )
    def test_draft_is_historical_continuation(self, tokens, max_k):
        """Any proposed draft must literally appear in the sequence right
        after an earlier occurrence of the tail bigram."""
        idx = NgramIndex(tokens)
        draft = idx.propose(tokens, max_k)
        assert len(draft) <= max_k
        if draft:
            a, b = tokens[-2], tokens[-1]
            found = False
            for i in range(1, len(tokens) - 1):
                if tokens[i - 1] == a and tokens[i] == b:
                    cont = tokens[i + 1 : i + 1 + len(draft)]
                    if cont == draft:
                        found = True
                        break
            assert found, (tokens, draft)

    @SETTINGS
    @given(
        tokens=st.lists(st.integers(0, 8), min_size=3, max_size=80),
        cut=st.integers(1, 79),
    )
    def test_incremental_equals_bulk(self, tokens, cut):
        cut = min(cut, len(tokens) - 1)
        inc = NgramIndex(tokens[:cut])
        inc.extend(tokens)
        bulk = NgramIndex(tokens)
        assert inc._map == bulk._map

    @SETTINGS
    @given(tokens=st.lists(st.integers(0, 8), min_size=3, max_size=60))
    def test_never_self_matches_tail(self, tokens):
        """The draft source must be a PREVIOUS occurrence — proposing the
        tail itself would draft the token being decoded."""
        idx = NgramIndex(tokens)
        draft = idx.propose(tokens, 4)
        # if the tail bigram occurs only once (at the very end), no draft
        a, b = tokens[-2], tokens[-1]
        occurrences = sum(
            1
            for i in range(1, len(tokens))
            if tokens[i - 1] == a and tokens[i] == b
        )
        if occurrences <= 1:
            assert draft == []


class TestFormGuideProperties:
    @SETTINGS
    @given(picks=st.lists(st.integers(0, 10**9), min_size=0, max_size=400))
    def test_any_walk_yields_valid_json(self, picks):
        """Walking a form by ANY sequence of allowed-token picks reaches
        done() with parseable JSON matching the declared schema."""
        from dts_amd.serving.structured import strategy_form
        from dts_amd.serving.tokenizer import SyntheticTokenizer

        tok = SyntheticTokenizer(512)
        guide = strategy_form(tok, 2)
        out = list(guide.initial_forced())  # skeleton prefix (engine.py:260)
        budget = guide.token_budget()
        steps = 0
        it = iter(picks)
        while not guide.done() and steps < budget + 64:
            steps += 1
            allowed = guide.allowed_tokens()
            if allowed is None:
                break
            choice = next(it, 7)
            t = allowed[choice % len(allowed)]
            out.append(t)
            out.extend(guide.on_token(t))
        assert guide.done(), "guide must terminate on any allowed walk"
        text = tok.decode([t for t in out if t < 256])
        full = guide.render(out) if hasattr(guide, "render") else None
        data = json.loads(full if full is not None else text)
        assert "nodes" in data and len(data["nodes"]) == 2


class TestChunkedPrefillProperties:
    @SETTINGS
    @given(
        prompt_len=st.integers(1, 300),
        budget=st.integers(4, 64),
        block_size=st.sampled_from([4, 8, 16]),
    )
    def test_chunks_cover_prompt_exactly(self, prompt_len, budget, block_size):
        """Chunked prefill schedules every prompt token exactly once and
        never exceeds the per-step token budget; the final chunk samples."""
        from dts_amd.llm.types import SamplingParams
        from dts_amd.serving.kv_cache import BlockManager
        from dts_amd.serving.scheduler import Scheduler
        from dts_amd.serving.sequence import Sequence

        sched = Scheduler(
            BlockManager(512, block_size), max_batch_tokens=budget
        )
        seq = Sequence(
            tokens=list(range(1, prompt_len + 1)),
            params=SamplingParams(max_tokens=4),
        )
        sched.add(seq)
        scheduled_tokens = 0
        steps = 0
        sampled = False
        while steps < prompt_len + 8:
            steps += 1
            b = sched.schedule()
            if b is None:
                break
            assert b.num_tokens <= budget
            scheduled_tokens += b.num_tokens
            sched.advance_computed(b)
            if any(s is seq for s in b._sampled_seqs):
                sampled = True
                break  # prompt fully processed; a token would be drawn
        assert sampled, "prefill never reached the sampling chunk"
        assert scheduled_tokens == prompt_len
