"""Direct coverage for the small parity utilities.

C8 (search/retry — ref backend/core/dts/retry.py:29-54), C19
(utils/config — ref backend/utils/config.py:7-131), C20 (utils/logging —
ref backend/utils/logging.py:10-35). These were previously only
exercised indirectly through evaluator/generator/server tests.
"""

import asyncio
import logging

import pytest

from dts_amd.llm.errors import TimeoutError_
from dts_amd.search.retry import llm_retry
from dts_amd.utils.config import Settings
from dts_amd.utils.logging import log_phase, logger


class TestRetry:
    def _run(self, coro):
        return asyncio.new_event_loop().run_until_complete(coro)

    def test_succeeds_after_transient_failures(self):
        calls = []

        @llm_retry(max_attempts=3, base_delay=0.001, max_delay=0.002)
        async def flaky():
            calls.append(1)
            if len(calls) < 3:
                raise TimeoutError_("transient")
            return "ok"

        assert self._run(flaky()) == "ok"
        assert len(calls) == 3

    def test_exhausted_attempts_reraise(self):
        calls = []

        @llm_retry(max_attempts=2, base_delay=0.001, max_delay=0.002)
        async def always_fails():
            calls.append(1)
            raise TimeoutError_("still down")

        with pytest.raises(TimeoutError_):
            self._run(always_fails())
        assert len(calls) == 2

    def test_non_retryable_raises_immediately(self):
        calls = []

        @llm_retry(max_attempts=3, base_delay=0.001)
        async def bad():
            calls.append(1)
            raise ValueError("programmer error")

        with pytest.raises(ValueError):
            self._run(bad())
        assert len(calls) == 1

    def test_preserves_function_identity(self):
        @llm_retry()
        async def named():  # pragma: no cover - only metadata checked
            return 1

        assert named.__name__ == "named"


class TestSettings:
    def test_defaults(self, monkeypatch):
        for var in ("DTS_MODEL", "DTS_PORT", "DTS_KV_FRACTION", "DTS_SCORING_MODE"):
            monkeypatch.delenv(var, raising=False)
        s = Settings()
        assert s.model_name == "llama-3-8b"
        assert s.server_port == 8000
        assert s.kv_memory_fraction == 0.75
        assert s.scoring_mode == "comparative"

    def test_env_overrides_and_casts(self, monkeypatch):
        monkeypatch.setenv("DTS_MODEL", "mixtral-8x7b")
        monkeypatch.setenv("DTS_PORT", "9001")
        monkeypatch.setenv("DTS_KV_FRACTION", "0.5")
        monkeypatch.setenv("DTS_BRANCHES", "4")
        s = Settings()
        assert s.model_name == "mixtral-8x7b"
        assert s.server_port == 9001
        assert s.kv_memory_fraction == 0.5
        assert s.init_branches == 4

    def test_bool_casting_helper(self, monkeypatch):
        from dts_amd.utils.config import _env

        monkeypatch.setenv("DTS_X_FLAG", "true")
        assert _env("DTS_X_FLAG", False, bool) is True
        monkeypatch.setenv("DTS_X_FLAG", "0")
        assert _env("DTS_X_FLAG", True, bool) is False
        monkeypatch.delenv("DTS_X_FLAG")
        assert _env("DTS_X_FLAG", True, bool) is True


class TestLogging:
    def test_singleton_no_duplicate_handlers(self):
        import dts_amd.utils.logging as L

        before = len(logger.handlers)
        again = L._build_logger()
        assert again is logger
        assert len(logger.handlers) == before

    def test_log_phase_format(self):
        # the singleton logger has propagate=False, so capture directly
        records = []

        class _Catch(logging.Handler):
            def emit(self, record):
                records.append(record)

        h = _Catch(level=logging.INFO)
        logger.addHandler(h)
        try:
            log_phase("EXPAND", "branch 3", indent=1)
        finally:
            logger.removeHandler(h)
        assert any(
            r.getMessage() == "[DTS:EXPAND]   branch 3" for r in records
        )
