"""Chat message / completion datatypes.

Parity: reference backend/llm/types.py:23-73 (Message/Usage/Completion),
re-designed as lightweight dataclasses — the hot path here is an in-process
GPU engine, not an HTTP client, so no pydantic validation cost per call.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional


@dataclass
class Message:
    """A chat message (role in {system,user,assistant})."""

    role: str
    content: Optional[str] = None

    @classmethod
    def system(cls, content: str) -> "Message":
        return cls(role="system", content=content)

    @classmethod
    def user(cls, content: str) -> "Message":
        return cls(role="user", content=content)

    @classmethod
    def assistant(cls, content: Optional[str] = None) -> "Message":
        return cls(role="assistant", content=content)

    def to_dict(self) -> dict:
        return {"role": self.role, "content": self.content}


@dataclass
class Usage:
    """Token usage for one completion (ref llm/types.py:50-56)."""

    prompt_tokens: int = 0
    completion_tokens: int = 0
    total_tokens: int = 0

    def __post_init__(self) -> None:
        if self.total_tokens == 0:
            self.total_tokens = self.prompt_tokens + self.completion_tokens


@dataclass
class Completion:
    """A completion response (ref llm/types.py:58-73).

    `data` carries the parsed JSON object when structured output was
    requested (ref client.py:141-203 semantics).
    """

    message: Message
    usage: Optional[Usage] = None
    model: Optional[str] = None
    finish_reason: Optional[str] = None
    data: Optional[dict] = None

    @property
    def content(self) -> Optional[str]:
        return self.message.content


@dataclass
class SamplingParams:
    """Sampling controls threaded from the search layer to the GPU sampler.

    The reference exposes only temperature (client.py:334-362); the local
    engine additionally needs explicit token budgets (the reference relies on
    the remote API's implicit limits — SURVEY.md §4.1.7) and seeds for
    reproducible tests.
    """

    temperature: float = 0.7
    top_p: float = 0.95
    top_k: int = 0  # 0 = disabled
    max_tokens: int = 256
    stop: list = field(default_factory=list)  # stop strings
    seed: Optional[int] = None
    json_mode: bool = False
