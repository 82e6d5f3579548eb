"""Flattened forward-batch metadata shared by scheduler and models.

One engine step runs ONE model forward over a flat token batch that mixes
prefill chunks and decode tokens (token-level continuous batching,
SURVEY.md §2.3 "Continuous-batching scheduler"). Prefill tokens come first,
then one decode token per running sequence.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class ForwardBatch:
    # flat inputs [T]
    token_ids: torch.Tensor
    positions: torch.Tensor
    slot_mapping: torch.Tensor  # flat cache slot per token
    # prefill section (first sum(q_lens) tokens of the batch)
    num_prefill_seqs: int = 0
    num_prefill_tokens: int = 0
    cu_q: Optional[torch.Tensor] = None  # [P+1] int32
    prefill_block_tables: Optional[torch.Tensor] = None  # [P, max_blocks]
    prefill_kv_lens: Optional[torch.Tensor] = None  # [P]
    # decode section
    num_decode_seqs: int = 0
    decode_block_tables: Optional[torch.Tensor] = None  # [D, max_blocks]
    decode_kv_lens: Optional[torch.Tensor] = None  # [D]
    # rows of the flat batch to compute logits for (last token of each
    # prefill chunk that completed its prompt + every decode token)
    sample_indices: Optional[torch.Tensor] = None

    @property
    def num_tokens(self) -> int:
        return int(self.token_ids.shape[0])

    def to(self, device) -> "ForwardBatch":
        def mv(t):
            return t.to(device, non_blocking=True) if t is not None else None

        return ForwardBatch(
            token_ids=mv(self.token_ids),
            positions=mv(self.positions),
            slot_mapping=mv(self.slot_mapping),
            num_prefill_seqs=self.num_prefill_seqs,
            num_prefill_tokens=self.num_prefill_tokens,
            cu_q=mv(self.cu_q),
            prefill_block_tables=mv(self.prefill_block_tables),
            prefill_kv_lens=mv(self.prefill_kv_lens),
            num_decode_seqs=self.num_decode_seqs,
            decode_block_tables=mv(self.decode_block_tables),
            decode_kv_lens=mv(self.decode_kv_lens),
            sample_indices=mv(self.sample_indices),
        )
