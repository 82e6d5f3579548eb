"""Inference backend protocol + the `LLM` front-end used by the search layer.

This is THE seam identified in SURVEY.md §3.5: the reference funnels every
call through `LLM.complete(messages, model, temperature, structured_output,
...) -> Completion` (ref backend/llm/client.py:78-203) over HTTPS; here the
same interface is backed by an in-process MI355X serving engine (or a fake
for tests). Kept semantics:

 - structured_output=True → parsed JSON in `Completion.data`, with
   <think>-tag stripping + markdown-fence extraction + bounded parse-retry
   (ref client.py:141-203, 453-478);
 - empty/invalid output surfaces as typed errors (ref errors.py), which the
   search layer's retry decorator handles.
"""

from __future__ import annotations

import json
import re
from typing import Optional, Protocol, runtime_checkable

from dts_amd.llm.errors import JSONParseError
from dts_amd.llm.types import Completion, Message, SamplingParams
from dts_amd.utils.logging import logger


@runtime_checkable
class InferenceBackend(Protocol):
    """Anything that can turn a chat into a completion.

    Implementations: serving.LocalBackend (MI355X engine), llm.FakeBackend
    (deterministic, for the reference-style mock test seam — SURVEY.md §4).
    """

    async def chat(
        self,
        messages: list[Message],
        params: SamplingParams,
        model: Optional[str] = None,
    ) -> Completion:
        ...


# Default generation budgets per structured-ness. The reference sets no
# max_tokens; a local engine must bound generation.
DEFAULT_MAX_TOKENS = 512
DEFAULT_JSON_MAX_TOKENS = 1024

_FENCE_RE = re.compile(r"```(?:json)?\s*(.*?)\s*```", re.DOTALL)
_THINK_RE = re.compile(r"<think>.*?</think>", re.DOTALL)


def strip_think_tags(text: str) -> str:
    """Drop <think>...</think> blocks (ref client.py:453-457)."""
    return _THINK_RE.sub("", text).strip()


def extract_json_object(text: str) -> str:
    """Pull a JSON object string out of model text (ref client.py:459-478).

    Handles raw JSON, markdown-fenced JSON, and JSON embedded in prose
    (first balanced {...} span).
    """
    text = text.strip()
    if text.startswith("{") and text.endswith("}"):
        return text
    m = _FENCE_RE.search(text)
    if m:
        return m.group(1)
    # first balanced object
    start = text.find("{")
    if start != -1:
        depth = 0
        in_str = False
        esc = False
        for i in range(start, len(text)):
            c = text[i]
            if in_str:
                if esc:
                    esc = False
                elif c == "\\":
                    esc = True
                elif c == '"':
                    in_str = False
                continue
            if c == '"':
                in_str = True
            elif c == "{":
                depth += 1
            elif c == "}":
                depth -= 1
                if depth == 0:
                    return text[start : i + 1]
    return text


def parse_json_completion(text: str) -> dict:
    """Parse model text into a JSON object, raising JSONParseError."""
    cleaned = strip_think_tags(text or "")
    candidate = extract_json_object(cleaned)
    try:
        obj = json.loads(candidate)
    except (json.JSONDecodeError, ValueError) as e:
        raise JSONParseError(f"invalid JSON from model: {e}", raw=text or "") from e
    if not isinstance(obj, dict):
        raise JSONParseError("model JSON is not an object", raw=text or "")
    return obj


class LLM:
    """Provider-agnostic completion front-end (ref client.py:35-203).

    One instance per search; `default_model` plays the role of the
    reference's `_default_model` (ref client.py:119-121 model fallback).
    """

    def __init__(
        self,
        backend: InferenceBackend,
        default_model: Optional[str] = None,
        max_json_retries: int = 3,
    ) -> None:
        self.backend = backend
        self._default_model = default_model
        self.max_json_retries = max_json_retries

    async def complete(
        self,
        messages: list[Message],
        model: Optional[str] = None,
        temperature: float = 0.7,
        structured_output: bool = False,
        max_tokens: Optional[int] = None,
        top_p: float = 0.95,
        seed: Optional[int] = None,
        **_ignored,
    ) -> Completion:
        """Complete a chat; with structured_output, retry until valid JSON.

        Mirrors ref client.py:78-203: up to `max_json_retries` attempts,
        each a fresh generation; on success `Completion.data` holds the
        parsed object.
        """
        model = model or self._default_model
        params = SamplingParams(
            temperature=temperature,
            top_p=top_p,
            max_tokens=max_tokens
            or (DEFAULT_JSON_MAX_TOKENS if structured_output else DEFAULT_MAX_TOKENS),
            seed=seed,
            json_mode=structured_output,
        )

        if not structured_output:
            completion = await self.backend.chat(messages, params, model=model)
            if completion.message.content:
                completion.message.content = strip_think_tags(completion.message.content)
            return completion

        last_err: Optional[Exception] = None
        for attempt in range(self.max_json_retries):  # ref client.py:148-203
            completion = await self.backend.chat(messages, params, model=model)
            try:
                completion.data = parse_json_completion(completion.message.content or "")
                return completion
            except JSONParseError as e:
                last_err = e
                logger.warning(
                    "JSON parse failed (attempt %d/%d): %s",
                    attempt + 1,
                    self.max_json_retries,
                    e,
                )
                # vary the seed so a deterministic sampler can escape
                if params.seed is not None:
                    params.seed += 1
        raise last_err  # type: ignore[misc]

    async def stream(
        self,
        messages: list,
        model: Optional[str] = None,
        temperature: float = 0.7,
        max_tokens: Optional[int] = None,
        top_p: float = 0.95,
        seed: Optional[int] = None,
        **_ignored,
    ):
        """Async iterator of text deltas (ref client.py:205-272). Backends
        without native streaming fall back to one whole-message chunk."""
        params = SamplingParams(
            temperature=temperature,
            top_p=top_p,
            max_tokens=max_tokens or DEFAULT_MAX_TOKENS,
            seed=seed,
        )
        model = model or self._default_model
        backend_stream = getattr(self.backend, "stream", None)
        if backend_stream is None:
            completion = await self.backend.chat(messages, params, model=model)
            if completion.message.content:
                yield strip_think_tags(completion.message.content)
            return
        async for delta in backend_stream(messages, params, model=model):
            yield delta
