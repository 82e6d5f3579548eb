"""Retry decorator + event emission utilities."""

import asyncio

import pytest

from dts_amd.llm.errors import BackendError, JSONParseError, LLMError
from dts_amd.search.events import create_event_emitter, emit_event
from dts_amd.search.retry import llm_retry


class TestRetry:
    def test_retries_transient_then_succeeds(self, run_async):
        calls = []

        @llm_retry(max_attempts=3, base_delay=0.01)
        async def flaky():
            calls.append(1)
            if len(calls) < 3:
                raise BackendError("transient")
            return "ok"

        assert run_async(flaky()) == "ok"
        assert len(calls) == 3

    def test_exhausts_and_raises(self, run_async):
        @llm_retry(max_attempts=2, base_delay=0.01)
        async def always_fails():
            raise JSONParseError("bad json")

        with pytest.raises(JSONParseError):
            run_async(always_fails())

    def test_non_retryable_passes_through(self, run_async):
        calls = []

        @llm_retry(max_attempts=3, base_delay=0.01)
        async def fatal():
            calls.append(1)
            raise ValueError("not transient")

        with pytest.raises(ValueError):
            run_async(fatal())
        assert len(calls) == 1  # no retries on non-LLM errors


class TestEvents:
    def test_async_callback_invoked(self, run_async):
        seen = []

        async def cb(t, d):
            seen.append((t, d))

        run_async(emit_event(cb, "x", {"a": 1}))
        assert seen == [("x", {"a": 1})]

    def test_sync_callback_and_errors_swallowed(self, run_async):
        def bad_cb(t, d):
            raise RuntimeError("boom")

        run_async(emit_event(bad_cb, "x", {}))  # must not raise

    def test_fire_and_forget_emitter(self, run_async):
        seen = []

        async def cb(t, d):
            seen.append(t)

        async def main():
            emit = create_event_emitter(cb)
            emit("e1", {})
            emit("e2", {})
            await asyncio.sleep(0)
            await asyncio.sleep(0)

        run_async(main())
        assert seen == ["e1", "e2"]

    def test_none_callback_noop(self, run_async):
        run_async(emit_event(None, "x", {}))
        create_event_emitter(None)("x", {})  # no loop needed, no-op
