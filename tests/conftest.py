"""Shared pytest fixtures.

Mirrors the reference's fixture strategy (ref tests/conftest.py:25-218):
canned messages/strategies/nodes plus a deterministic backend in place of
the reference's AsyncMock LLM.
"""

import asyncio
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")
    config.addinivalue_line("markers", "dist: spawns torch.distributed processes (gloo)")


@pytest.fixture
def run_async():
    """Run a coroutine to completion on a fresh event loop."""

    def _run(coro):
        return asyncio.run(coro)

    return _run


@pytest.fixture
def fake_llm():
    from dts_amd.llm import LLM, FakeBackend

    backend = FakeBackend()
    return LLM(backend, default_model="fake-model")


@pytest.fixture
def dts_config():
    from dts_amd.search import DTSConfig

    return DTSConfig(
        goal="Help the user understand gradient descent",
        first_message="Can you explain how training works?",
        init_branches=3,
        turns_per_branch=2,
        user_intents_per_branch=2,
        scoring_mode="absolute",
        prune_threshold=5.0,
        seed=1234,
    )
