"""Error taxonomy for the inference layer.

Parity: reference backend/llm/errors.py:1-69. HTTP-status-specific classes
(RateLimit/Authentication/Server) have no local analogue; the retained
classes are the ones the search layer's retry policy keys on
(ref core/dts/retry.py:29-54) plus engine-local failure modes.
"""

from __future__ import annotations


class LLMError(Exception):
    """Base class for all inference-layer errors."""


class BackendError(LLMError):
    """The serving engine failed internally (kernel error, OOM, ...)."""


class TimeoutError_(LLMError):
    """A generation request exceeded its deadline."""


class ContextLengthError(LLMError):
    """Prompt + generation exceeds the model's max context."""


class JSONParseError(LLMError):
    """Structured output requested but no valid JSON could be extracted."""

    def __init__(self, message: str, raw: str = "") -> None:
        super().__init__(message)
        self.raw = raw


class EmptyResponseError(LLMError):
    """Model produced an empty completion (ref simulator.py:28-31)."""


#: Errors considered transient — the search layer retries on these
#: (ref core/dts/retry.py:36-44 lists RateLimit/Server/Timeout/Connection/JSONParse).
RETRYABLE_ERRORS = (TimeoutError_, JSONParseError, BackendError)
