"""Adapter exposing the C++ CoreScheduler behind the Python Scheduler API.

The native core (dts_amd/core/csrc/core.cpp) owns token/block/prefix-cache
state and emits each step's flat batch as ready-made torch tensors; the
Python `Sequence` objects remain the carriers of sampling params, guides
and output bookkeeping. Token appends flow through this adapter so both
sides stay in sync.
"""

from __future__ import annotations

from typing import Optional

from dts_amd.serving.batch import ForwardBatch
from dts_amd.serving.sequence import Sequence, SeqStatus


class NativeScheduler:
    def __init__(
        self,
        num_blocks: int,
        block_size: int,
        max_batch_tokens: int = 8192,
        max_running: int = 256,
        spec_k: int = 0,
        max_spec_rows: int = 16,
    ) -> None:
        from dts_amd.core import load_core

        core_mod = load_core()
        if core_mod is None:
            raise RuntimeError("_dts_core extension not built")
        self.core = core_mod.CoreScheduler(
            num_blocks,
            block_size,
            max_batch_tokens,
            max_running,
            spec_k,
            max_spec_rows,
        )
        self.block_size = block_size
        self._seqs: dict = {}
        self.stuck: list = []

    # -- stats shim (engine.cache_stats reads these off block_manager) ----
    @property
    def cache_hit_tokens(self) -> int:
        return self.core.cache_hit_tokens

    @property
    def preemptions(self) -> int:
        return self.core.preemptions

    @property
    def cache_miss_tokens(self) -> int:
        return self.core.cache_miss_tokens

    def num_free(self) -> int:
        return self.core.num_free()

    # -- scheduler API -----------------------------------------------------
    def add(self, seq: Sequence) -> None:
        self._seqs[seq.seq_id] = seq
        seq.status = SeqStatus.WAITING
        # guided (constrained-form) seqs never speculate: drafts would
        # fight the forced-token machinery
        self.core.add(
            seq.seq_id, [int(t) for t in seq.tokens], seq.guide is None
        )

    def set_accepted(self, seq: Sequence, accepted: int) -> None:
        """Engine feedback after speculative verification."""
        self.core.set_sched_chunk(seq.seq_id, int(accepted))

    # -- chained-decode hooks (serving/chain.py) -----------------------
    def waiting_count(self) -> int:
        return self.core.waiting_count()

    def reserve_tokens(self, seq: Sequence, upto_tokens: int) -> bool:
        return self.core.reserve_tokens(seq.seq_id, int(upto_tokens))

    def chain_advance(self, seq: Sequence, tok: int) -> None:
        seq.append_token(tok)
        self.core.chain_advance(seq.seq_id, int(tok))

    def block_table_of(self, seq: Sequence) -> list:
        return list(self.core.get_block_table(seq.seq_id))

    def num_computed_of(self, seq: Sequence) -> int:
        return int(self.core.num_computed(seq.seq_id))

    def append_token(self, seq: Sequence, tok: int) -> None:
        seq.append_token(tok)
        self.core.append_token(seq.seq_id, int(tok))

    def extend_tokens(self, seq: Sequence, toks: list) -> None:
        seq.tokens.extend(toks)
        seq.output_tokens.extend(toks)
        self.core.extend_tokens(seq.seq_id, [int(t) for t in toks])

    def has_work(self) -> bool:
        return self.core.has_work()

    @property
    def running(self) -> list:
        # used only on error paths — approximate with all live seqs
        return [s for s in self._seqs.values() if s.status == SeqStatus.RUNNING]

    def schedule(self) -> Optional[ForwardBatch]:
        d = self.core.schedule()
        for sid in self.core.take_stuck():
            # stuck = terminal; drop our entry too or it leaks (the engine
            # fails the future without calling finish/abort)
            seq = self._seqs.pop(sid, None)
            if seq is not None:
                self.stuck.append(seq)
        if not d:
            return None
        scheduled = [self._seqs[i] for i in d["scheduled_ids"]]
        sampled = [self._seqs[i] for i in d["sampled_ids"]]
        for s in scheduled:
            s.status = SeqStatus.RUNNING
        batch = ForwardBatch(
            token_ids=d["token_ids"],
            positions=d["positions"],
            slot_mapping=d["slot_mapping"],
            num_prefill_seqs=d["num_prefill_seqs"],
            num_prefill_tokens=d["num_prefill_tokens"],
            cu_q=d.get("cu_q"),
            prefill_block_tables=d.get("prefill_block_tables"),
            prefill_kv_lens=d.get("prefill_kv_lens"),
            num_decode_seqs=d["num_decode_seqs"],
            decode_block_tables=d.get("decode_block_tables"),
            decode_kv_lens=d.get("decode_kv_lens"),
            sample_indices=d["sample_indices"],
        )
        batch._scheduled = scheduled  # type: ignore[attr-defined]
        batch._sampled_seqs = sampled  # type: ignore[attr-defined]
        batch._sample_pos = d["sample_pos"]  # type: ignore[attr-defined]
        batch._row_groups = [  # type: ignore[attr-defined]
            (self._seqs[sid], n) for sid, n in d["row_groups"]
        ]
        batch._spec_drafts = {  # type: ignore[attr-defined]
            sid: [int(t) for t in toks] for sid, toks in d["spec"]
        }
        return batch

    def advance_computed(self, batch: ForwardBatch) -> None:
        self.core.advance()

    def finish(self, seq: Sequence, reason: str) -> None:
        seq.status = SeqStatus.FINISHED
        seq.finish_reason = reason
        self.core.finish(seq.seq_id)
        self._seqs.pop(seq.seq_id, None)

    def abort(self, seq: Sequence) -> None:
        seq.status = SeqStatus.ABORTED
        self.core.abort(seq.seq_id)
        self._seqs.pop(seq.seq_id, None)
