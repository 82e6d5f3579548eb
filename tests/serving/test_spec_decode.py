"""Prompt-lookup speculative decoding (serving/spec.py + both schedulers).

The invariant under test everywhere: speculation must be OUTPUT-INVISIBLE
— exact-match verification emits the model's own samples, so any request
produces the identical token stream with spec on or off; speculation only
changes how many engine steps that stream takes.
"""

import pytest
import torch

from dts_amd.llm.types import SamplingParams
from dts_amd.serving.kv_cache import BlockManager
from dts_amd.serving.scheduler import Scheduler
from dts_amd.serving.sequence import Sequence
from dts_amd.serving.spec import NgramIndex


class TestNgramIndex:
    def test_propose_from_repetition(self):
        toks = [1, 2, 3, 4, 5, 1, 2]
        ix = NgramIndex(toks)
        # last bigram (1,2) occurred at index 0-1 → continuation 3,4,5...
        assert ix.propose(toks, 3) == [3, 4, 5]
        assert ix.propose(toks, 2) == [3, 4]

    def test_no_match(self):
        toks = [1, 2, 3, 4, 5]
        ix = NgramIndex(toks)
        assert ix.propose(toks, 4) == []

    def test_latest_occurrence_wins(self):
        # (1,2) appears twice before the tail; draft should follow the
        # most recent previous occurrence
        toks = [1, 2, 7, 7, 1, 2, 8, 8, 1, 2]
        ix = NgramIndex(toks)
        assert ix.propose(toks, 2) == [8, 8]

    def test_incremental_extend_matches_bulk(self):
        toks = [3, 1, 4, 1, 5, 9, 2, 6, 5, 3, 5, 1, 4]
        bulk = NgramIndex(toks)
        inc = NgramIndex()
        live: list = []
        for t in toks:
            live.append(t)
            inc.extend(live)
        assert inc.propose(live, 4) == bulk.propose(toks, 4)

    def test_short_sequences(self):
        assert NgramIndex([1]).propose([1], 4) == []
        assert NgramIndex([1, 2]).propose([1, 2], 4) == []


def _mk_scheduler(spec_k=4, num_blocks=64, block_size=4):
    return Scheduler(
        BlockManager(num_blocks, block_size),
        max_batch_tokens=256,
        spec_k=spec_k,
    )


def _decode_state(sched, seq):
    """Drive seq through prefill so its next schedule is a decode row."""
    sched.add(seq)
    b = sched.schedule()
    assert b is not None
    sched.advance_computed(b)
    return b


class TestSchedulerSpecRows:
    def test_draft_rows_emitted(self):
        sched = _mk_scheduler()
        # prompt with a repeated bigram so the tail (1,2) has history
        seq = Sequence(
            tokens=[1, 2, 3, 4, 5, 1, 2], params=SamplingParams(max_tokens=16)
        )
        _decode_state(sched, seq)
        # simulate the engine sampling a token: now it's a decode seq
        sched.append_token(seq, 9)
        b = sched.schedule()
        # tail bigram is (2, 9) — no match; append tokens to recreate one
        sched.set_accepted(seq, 1)
        sched.advance_computed(b)
        sched.append_token(seq, 1)
        b = sched.schedule()
        sched.set_accepted(seq, 1)
        sched.advance_computed(b)
        sched.append_token(seq, 2)
        # tail bigram (1,2) matches twice-earlier occurrence → drafts
        b = sched.schedule()
        assert b.num_decode_seqs > 1  # rows, not seqs
        (s, n_rows) = b._row_groups[-1]
        assert s is seq and n_rows == 1 + len(b._spec_drafts[seq.seq_id])
        drafts = b._spec_drafts[seq.seq_id]
        # tokens are now [1,2,3,4,5,1,2,9,1,2]; the latest previous (1,2)
        # ends at index 6, so the draft continues with [9, 1]
        assert drafts[:2] == [9, 1]
        # row j: position p+j, kv_len p+j+1, sample_pos len+j
        p = seq.num_computed
        L = len(seq.tokens)
        for j in range(n_rows):
            assert int(b.positions[j]) == p + j
            assert int(b.decode_kv_lens[j]) == p + j + 1
            assert b._sample_pos[j] == L + j
        # token ids: tail token then the drafts
        assert int(b.token_ids[0]) == seq.tokens[p]
        assert [int(t) for t in b.token_ids[1:n_rows]] == drafts

    def test_accepted_advance(self):
        sched = _mk_scheduler()
        seq = Sequence(
            tokens=[1, 2, 3, 4, 1, 2], params=SamplingParams(max_tokens=16)
        )
        _decode_state(sched, seq)
        sched.append_token(seq, 3)  # tail (2,3): continuation 4 exists
        b = sched.schedule()
        assert seq.seq_id in b._spec_drafts
        drafts = b._spec_drafts[seq.seq_id]
        # engine accepts all drafts + 1: appends 1+len(drafts) tokens
        for t in [4] + [77] * len(drafts):
            sched.append_token(seq, t)
        sched.set_accepted(seq, 1 + len(drafts))
        sched.advance_computed(b)
        assert seq.num_computed == len(seq.tokens) - 1  # decode invariant

    def test_no_spec_for_guided(self):
        sched = _mk_scheduler()
        seq = Sequence(
            tokens=[1, 2, 3, 1, 2], params=SamplingParams(max_tokens=16)
        )
        seq.guide = object()  # any guide disables speculation
        sched.add(seq)
        b = sched.schedule()
        sched.advance_computed(b)
        sched.append_token(seq, 3)
        b = sched.schedule()
        assert b.num_decode_seqs == 1
        assert not b._spec_drafts

    def test_budget_starved_chunk1_is_prefill_not_decode(self):
        """A mid-prompt chunk of 1 (batch budget exhausted) must not be
        sampled as if it were a decode row."""
        sched = Scheduler(BlockManager(64, 4), max_batch_tokens=5, spec_k=0)
        a = Sequence(tokens=[1, 2, 3, 4], params=SamplingParams())
        z = Sequence(tokens=[5, 6, 7, 8], params=SamplingParams())
        sched.add(a)
        sched.add(z)
        b = sched.schedule()
        # a takes 4 budget, z gets chunk 1 of its 4-token prompt
        assert b.num_prefill_seqs == 2
        assert b.num_decode_seqs == 0
        # only a (whose chunk completes its prompt) samples
        assert b._sampled_seqs == [a]


@pytest.mark.parametrize("use_native", [False, True])
class TestEngineSpecEquivalence:
    def _engine(self, use_native, spec_k, monkeypatch):
        if use_native:
            from dts_amd.core import load_core

            if load_core() is None:
                pytest.skip("native core not built")
            monkeypatch.setenv("DTS_NATIVE_CORE", "1")
        else:
            monkeypatch.setenv("DTS_NATIVE_CORE", "0")
        from dts_amd.serving.engine import ServingEngine

        return ServingEngine(
            model_name="llama-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=512,
            block_size=8,
            weight_seed=11,
            spec_k=spec_k,
        )

    def _gen(self, eng, prompt, seed, temperature, n=48):
        fut = eng.submit_tokens(
            list(prompt),
            SamplingParams(max_tokens=n, seed=seed, temperature=temperature),
        )
        eng.run_until_idle()
        return fut.result(timeout=60).token_ids

    def test_greedy_identical_and_accepts(self, use_native, monkeypatch):
        # greedy decode of a tiny random model loops quickly — the loop
        # IS the n-gram hit, so acceptance must kick in
        prompt = [300 + i for i in range(8)] * 2  # repetitive prompt
        eng_off = self._engine(use_native, 0, monkeypatch)
        ref = self._gen(eng_off, prompt, seed=None, temperature=0.0)
        eng_on = self._engine(use_native, 4, monkeypatch)
        out = self._gen(eng_on, prompt, seed=None, temperature=0.0)
        assert out == ref
        assert eng_on.spec_draft_tokens > 0
        assert eng_on.spec_accepted_tokens > 0
        # fewer steps is the whole point
        assert eng_on.steps < eng_off.steps

    def test_seeded_sampling_identical(self, use_native, monkeypatch):
        # near-greedy temperature: the tiny random model loops, so drafts
        # fire, while the seeded multinomial path (stateless per-position
        # generators) is what's being exercised
        prompt = [310, 311, 312, 310, 311, 312, 310, 311]
        eng_off = self._engine(use_native, 0, monkeypatch)
        ref = self._gen(eng_off, prompt, seed=1234, temperature=0.05)
        eng_on = self._engine(use_native, 4, monkeypatch)
        out = self._gen(eng_on, prompt, seed=1234, temperature=0.05)
        assert out == ref
        assert eng_on.spec_draft_tokens > 0

    def test_seeded_high_temp_identical(self, use_native, monkeypatch):
        # at T=0.7 drafts rarely fire on random weights, but when they
        # do the stream must still be identical — and the seeded stream
        # itself must match spec-off exactly
        prompt = [310, 311, 312, 310, 311, 312, 310, 311]
        eng_off = self._engine(use_native, 0, monkeypatch)
        ref = self._gen(eng_off, prompt, seed=77, temperature=0.7)
        eng_on = self._engine(use_native, 4, monkeypatch)
        out = self._gen(eng_on, prompt, seed=77, temperature=0.7)
        assert out == ref

    def test_stats_exposed(self, use_native, monkeypatch):
        eng = self._engine(use_native, 4, monkeypatch)
        stats = eng.cache_stats
        assert "spec_draft_tokens" in stats and "spec_accepted_tokens" in stats


class TestSchedulerDifferential:
    """Python and native schedulers must emit identical spec batches."""

    def test_same_drafts_both_schedulers(self):
        from dts_amd.core import load_core

        if load_core() is None:
            pytest.skip("native core not built")
        from dts_amd.serving.native_scheduler import NativeScheduler

        toks = [1, 2, 3, 4, 5, 1, 2]
        results = []
        for sched in (
            _mk_scheduler(spec_k=4),
            NativeScheduler(64, 4, max_batch_tokens=256, spec_k=4),
        ):
            seq = Sequence(
                tokens=list(toks), params=SamplingParams(max_tokens=16)
            )
            sched.add(seq)
            b = sched.schedule()
            sched.advance_computed(b)
            for t in (9, 1, 2):
                sched.append_token(seq, t)
                b = sched.schedule()
                sched.set_accepted(seq, 1)
                sched.advance_computed(b)
                last = b
            results.append(
                (
                    last.num_decode_seqs,
                    last._spec_drafts.get(seq.seq_id, []),
                    [int(x) for x in last.token_ids],
                    [int(x) for x in last.positions],
                    [int(x) for x in last.decode_kv_lens],
                    list(last._sample_pos),
                )
            )
        assert results[0] == results[1]


class TestSpecUnderPressure:
    """Drafts must not break KV accounting under pool pressure: draft
    rows allocate blocks ahead of acceptance, preemption recomputes, and
    the stream must still equal the spec-off run exactly."""

    @pytest.mark.parametrize("use_native", [False, True])
    def test_spec_with_preemption_matches(self, use_native, monkeypatch):
        if use_native:
            from dts_amd.core import load_core

            if load_core() is None:
                pytest.skip("native core not built")
        monkeypatch.setenv("DTS_NATIVE_CORE", "1" if use_native else "0")
        from dts_amd.serving.engine import ServingEngine

        def run(spec_k):
            eng = ServingEngine(
                model_name="llama-tiny",
                device="cpu",
                dtype=torch.float32,
                num_blocks=48,  # tight pool: forces preemption
                block_size=4,
                weight_seed=11,
                spec_k=spec_k,
            )
            # repetitive prompts → drafts propose; several seqs compete
            prompts = [
                [300 + (i % 6) for i in range(24)],
                [400 + (i % 5) for i in range(20)],
                [500 + (i % 4) for i in range(16)],
            ]
            futs = [
                eng.submit_tokens(
                    list(p),
                    SamplingParams(max_tokens=24, temperature=0.0, seed=None),
                )
                for p in prompts
            ]
            eng.run_until_idle()
            outs = [f.result(timeout=60).token_ids for f in futs]
            free = eng.scheduler.num_free() if hasattr(
                eng.scheduler, "num_free") else eng.block_manager.num_free()
            return outs, eng.spec_draft_tokens, free

        ref, _, _ = run(0)
        out, drafted, free = run(4)
        assert out == ref
        assert drafted > 0
        assert free == 47  # whole pool drained (1 block is scratch-reserved)


class TestSpecEquivalenceFuzz:
    """Randomized repetitive prompts: spec-on and spec-off streams must
    be identical for any (motif, temp, seed, k). A 50-trial sweep of
    this generator passed with 28 draft-accepting trials; these seeds
    are the committed regression subset."""

    @pytest.mark.parametrize("use_native", [False, True])
    @pytest.mark.parametrize("trial", [0, 3, 7, 11, 19])
    def test_random_trial(self, use_native, trial, monkeypatch):
        import random

        if use_native:
            from dts_amd.core import load_core

            if load_core() is None:
                pytest.skip("native core not built")
            monkeypatch.setenv("DTS_NATIVE_CORE", "1")
        else:
            monkeypatch.setenv("DTS_NATIVE_CORE", "0")
        from dts_amd.serving.engine import ServingEngine

        def make(spec_k):
            return ServingEngine(
                model_name="llama-tiny",
                device="cpu",
                dtype=torch.float32,
                num_blocks=512,
                block_size=8,
                weight_seed=11,
                spec_k=spec_k,
            )

        def gen(eng, prompt, seed, temperature, n):
            fut = eng.submit_tokens(
                list(prompt),
                SamplingParams(max_tokens=n, seed=seed, temperature=temperature),
            )
            eng.run_until_idle()
            return fut.result(timeout=60).token_ids

        rng = random.Random(trial)
        motif = [rng.randrange(250, 400) for _ in range(rng.randrange(3, 9))]
        prompt = (motif * rng.randrange(2, 5))[: rng.randrange(8, 30)]
        temp = rng.choice([0.0, 0.0, 0.05, 0.3, 0.7])
        seed = rng.randrange(10_000) if temp > 0 else None
        n = rng.randrange(16, 64)
        k = rng.choice([2, 4, 8])

        off = make(0)
        ref = gen(off, prompt, seed, temp, n)
        off.stop()
        on = make(k)
        out = gen(on, prompt, seed, temp, n)
        on.stop()
        assert out == ref
