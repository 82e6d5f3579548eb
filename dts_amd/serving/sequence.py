"""Sequence state for the continuous-batching scheduler."""

from __future__ import annotations

import itertools
from dataclasses import dataclass, field
from enum import Enum
from typing import Optional

from dts_amd.llm.types import SamplingParams

_seq_counter = itertools.count()


class SeqStatus(str, Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"
    ABORTED = "aborted"


@dataclass
class Sequence:
    """One generation request flowing through the engine.

    tokens = prompt + generated; `num_computed` counts tokens whose KV is in
    cache (advanced by prefix hits and by scheduled prefill chunks).
    """

    tokens: list
    params: SamplingParams
    seq_id: int = field(default_factory=lambda: next(_seq_counter))
    status: SeqStatus = SeqStatus.WAITING
    block_table: list = field(default_factory=list)
    num_computed: int = 0
    num_prompt_tokens: int = 0
    # decoding state
    output_tokens: list = field(default_factory=list)
    finish_reason: Optional[str] = None
    # constrained decoding hook (serving/structured.py); None = free sampling
    guide: Optional[object] = None
    # last full-block chain hash (for registering new full blocks)
    last_block_hash: int = 0
    num_hashed_blocks: int = 0
    arrival_order: int = 0
    # scheduled in a forward that has not been postprocessed yet
    in_flight: bool = False
    # optional per-token callback (called from the engine thread)
    stream_cb: Optional[object] = None

    def __post_init__(self) -> None:
        self.num_prompt_tokens = len(self.tokens)

    def __len__(self) -> int:
        return len(self.tokens)

    @property
    def num_generated(self) -> int:
        return len(self.output_tokens)

    def append_token(self, token_id: int) -> None:
        self.tokens.append(token_id)
        self.output_tokens.append(token_id)

    def blocks_needed(self, block_size: int, extra_tokens: int = 0) -> int:
        total = len(self.tokens) + extra_tokens
        return (total + block_size - 1) // block_size
