"""Fire-and-forget event emission helpers.

Parity: reference backend/core/dts/utils.py:53-102 (emit_event /
create_event_emitter): events must never block or crash the search loop.
"""

from __future__ import annotations

import asyncio
from typing import Callable, Optional

from dts_amd.utils.logging import logger


async def emit_event(
    callback: Optional[Callable],
    event_type: str,
    data: dict,
) -> None:
    if callback is None:
        return
    try:
        result = callback(event_type, data)
        if asyncio.iscoroutine(result):
            await result
    except Exception as e:  # noqa: BLE001 — events must never propagate
        logger.warning("Event callback error: %s", e)


def create_event_emitter(callback: Optional[Callable]) -> Callable[[str, dict], None]:
    def emit(event_type: str, data: dict) -> None:
        if callback is not None:
            asyncio.create_task(emit_event(callback, event_type, data))

    return emit
