"""Environment-driven settings.

Parity: reference backend/utils/config.py:7-131 (pydantic-settings from
.env: model names, timeouts, researcher knobs, server host/port). The
remote-API key fields have no local meaning; the retained surface is what
the server and engine factories read. Implemented with pydantic
BaseSettings when available, plain-env fallback otherwise.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field


def _env(name: str, default, cast=str):
    raw = os.environ.get(name)
    if raw is None:
        return default
    if cast is bool:
        return raw.lower() in ("1", "true", "yes", "on")
    return cast(raw)


@dataclass
class Settings:
    # model serving
    model_name: str = field(default_factory=lambda: _env("DTS_MODEL", "llama-3-8b"))
    device: str = field(default_factory=lambda: _env("DTS_DEVICE", "auto"))
    kv_memory_fraction: float = field(
        default_factory=lambda: _env("DTS_KV_FRACTION", 0.75, float)
    )
    max_batch_tokens: int = field(
        default_factory=lambda: _env("DTS_MAX_BATCH_TOKENS", 16384, int)
    )
    # search defaults (ref config.py knobs)
    init_branches: int = field(default_factory=lambda: _env("DTS_BRANCHES", 6, int))
    turns_per_branch: int = field(default_factory=lambda: _env("DTS_TURNS", 5, int))
    prune_threshold: float = field(
        default_factory=lambda: _env("DTS_PRUNE_THRESHOLD", 6.5, float)
    )
    scoring_mode: str = field(
        default_factory=lambda: _env("DTS_SCORING_MODE", "comparative")
    )
    # research (ref gpt-researcher knobs)
    research_cache_dir: str = field(
        default_factory=lambda: _env("DTS_RESEARCH_CACHE", ".cache/research")
    )
    research_provider: str = field(
        default_factory=lambda: _env("DTS_RESEARCH_PROVIDER", "auto")
    )
    # server
    server_host: str = field(default_factory=lambda: _env("DTS_HOST", "0.0.0.0"))
    server_port: int = field(default_factory=lambda: _env("DTS_PORT", 8000, int))
    log_level: str = field(default_factory=lambda: _env("DTS_LOG_LEVEL", "INFO"))


settings = Settings()
