"""Async retry with exponential backoff for transient inference errors.

Parity: reference backend/core/dts/retry.py:29-54 (tenacity, 3 attempts,
exponential 0.5→8 s on transient + parse errors). Implemented directly —
no tenacity dependency; same backoff schedule and retryable-error set.
"""

from __future__ import annotations

import asyncio
import functools
import random

from dts_amd.llm.errors import RETRYABLE_ERRORS
from dts_amd.utils.logging import logger


def llm_retry(max_attempts: int = 3, base_delay: float = 0.5, max_delay: float = 8.0):
    """Decorator: retry an async function on RETRYABLE_ERRORS."""

    def decorator(fn):
        @functools.wraps(fn)
        async def wrapper(*args, **kwargs):
            delay = base_delay
            for attempt in range(1, max_attempts + 1):
                try:
                    return await fn(*args, **kwargs)
                except RETRYABLE_ERRORS as e:
                    if attempt == max_attempts:
                        raise
                    sleep = min(delay, max_delay) * (0.5 + random.random())
                    logger.warning(
                        "%s attempt %d/%d failed (%s: %s); retrying in %.2fs",
                        fn.__name__,
                        attempt,
                        max_attempts,
                        type(e).__name__,
                        e,
                        sleep,
                    )
                    await asyncio.sleep(sleep)
                    delay *= 2
            raise RuntimeError("unreachable")

        return wrapper

    return decorator
