"""FastAPI WS/REST server.

Parity: reference backend/api/server.py:26-247 — WS endpoint speaking the
same protocol ({"type":"start_search","config":{...}} in, event stream
out, ping→pong), /health, /config defaults and /api/models. /api/models
lists the locally-served model registry instead of fetching OpenRouter
(there is no remote provider). The engine behind the socket is the
in-process MI355X serving stack; `create_app` receives a backend factory
so tests inject fakes exactly like the reference's patched
run_dts_session (ref tests/api/test_server.py:253-271).
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Callable, Optional

from fastapi import FastAPI, WebSocket, WebSocketDisconnect
from fastapi.middleware.cors import CORSMiddleware
from fastapi.responses import HTMLResponse

from dts_amd.llm.backend import LLM
from dts_amd.models.config import MODEL_REGISTRY
from dts_amd.server import service as dts_service
from dts_amd.server.schemas import SearchRequest
from dts_amd.utils.logging import logger

from dts_amd.utils.config import settings as _settings

# /config defaults reflect the env-driven Settings (ref utils/config.py
# pydantic-settings semantics: .env overrides, ref server.py:156-169)
DEFAULT_CONFIG = {
    "init_branches": _settings.init_branches,
    "turns_per_branch": _settings.turns_per_branch,
    "user_intents_per_branch": 3,
    "scoring_mode": _settings.scoring_mode,
    "prune_threshold": _settings.prune_threshold,
    "rounds": 1,
    "deep_research": False,
    "user_variability": False,
    "reasoning_enabled": False,
}


class ConnectionManager:
    """Tracks live websockets (ref server.py:38-59)."""

    def __init__(self) -> None:
        self.active: list = []

    async def connect(self, websocket: WebSocket) -> None:
        await websocket.accept()
        self.active.append(websocket)

    def disconnect(self, websocket: WebSocket) -> None:
        if websocket in self.active:
            self.active.remove(websocket)

    async def send_json(self, websocket: WebSocket, data: dict) -> None:
        await websocket.send_text(json.dumps(data))


def create_app(llm_factory: Optional[Callable[[], LLM]] = None) -> FastAPI:
    app = FastAPI(title="dts_amd", version="0.1.0")
    app.add_middleware(
        CORSMiddleware,
        allow_origins=["*"],
        allow_credentials=True,
        allow_methods=["*"],
        allow_headers=["*"],
    )
    manager = ConnectionManager()
    app.state.llm_factory = llm_factory
    app.state.llm = None

    def get_llm() -> LLM:
        if app.state.llm is None:
            factory = app.state.llm_factory
            if factory is None:
                raise RuntimeError(
                    "no inference backend configured; start via "
                    "`python -m dts_amd.server` or pass llm_factory"
                )
            app.state.llm = factory()
        return app.state.llm

    # ------------------------------------------------------------------
    @app.websocket("/ws")
    async def websocket_endpoint(websocket: WebSocket):
        await manager.connect(websocket)
        try:
            while True:
                raw = await websocket.receive_text()
                try:
                    message = json.loads(raw)
                except json.JSONDecodeError:
                    await manager.send_json(
                        websocket,
                        {"type": "error", "data": {"message": "invalid JSON"}},
                    )
                    continue
                mtype = message.get("type")
                if mtype == "ping":
                    await manager.send_json(websocket, {"type": "pong", "data": {}})
                elif mtype == "start_search":
                    await handle_search(websocket, message.get("config", {}))
                else:
                    await manager.send_json(
                        websocket,
                        {
                            "type": "error",
                            "data": {"message": f"unknown message type: {mtype}"},
                        },
                    )
        except WebSocketDisconnect:
            manager.disconnect(websocket)

    async def handle_search(websocket: WebSocket, config: dict) -> None:
        try:
            request = SearchRequest(**config)
        except Exception as e:  # noqa: BLE001 — validation error to client
            await manager.send_json(
                websocket,
                {"type": "error", "data": {"message": f"invalid config: {e}"}},
            )
            return
        try:
            async for event in dts_service.run_dts_session(request, get_llm()):
                await manager.send_json(websocket, event)
        except WebSocketDisconnect:
            manager.disconnect(websocket)
        except Exception as e:  # noqa: BLE001
            logger.exception("search failed: %s", e)
            await manager.send_json(
                websocket, {"type": "error", "data": {"message": str(e)}}
            )

    # ------------------------------------------------------------------
    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/config")
    async def config():
        return DEFAULT_CONFIG

    @app.get("/api/models")
    async def models():
        """Locally-served models (replaces the OpenRouter list fetch,
        ref server.py:172-232)."""
        out = []
        for name, spec in MODEL_REGISTRY.items():
            out.append(
                {
                    "id": name,
                    "name": name,
                    "context_length": spec.max_position,
                    "architecture": {"modality": "text->text"},
                    "pricing": {"prompt": "0", "completion": "0"},
                    "local": True,
                }
            )
        return {"data": out}

    @app.get("/metrics")
    async def metrics():
        """Prometheus metrics (framework addition — the reference has no
        metrics endpoint, SURVEY.md §5 Tracing: observability is logs
        only). Exposes the live serving engines' counters."""
        from fastapi.responses import PlainTextResponse
        from prometheus_client import (
            CollectorRegistry,
            Gauge,
            generate_latest,
        )

        registry = CollectorRegistry()
        g = Gauge(
            "dts_engine_stat",
            "dts_amd serving-engine counter",
            ["model", "stat"],
            registry=registry,
        )
        llm = app.state.llm
        backend = getattr(llm, "backend", None) if llm else None
        engines = getattr(backend, "engines", None) or {}
        for name, eng in engines.items():
            try:
                for stat, value in eng.cache_stats.items():
                    if isinstance(value, bool):
                        value = int(value)
                    if isinstance(value, (int, float)):
                        g.labels(model=name, stat=stat).set(value)
            except Exception:  # noqa: BLE001 — metrics must never 500
                continue
        return PlainTextResponse(
            generate_latest(registry), media_type="text/plain; version=0.0.4"
        )

    @app.get("/", response_class=HTMLResponse)
    async def index():
        static_index = Path(__file__).parent / "static" / "index.html"
        if static_index.exists():
            return HTMLResponse(static_index.read_text())
        return HTMLResponse(
            "<html><body><h3>dts_amd server</h3>"
            "<p>WS endpoint: /ws — protocol-compatible with the DTS "
            "frontend.</p></body></html>"
        )

    return app


def main() -> None:
    import argparse

    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--host", default=_settings.server_host)
    p.add_argument("--port", type=int, default=_settings.server_port)
    p.add_argument("--model", default=_settings.model_name)
    p.add_argument("--device", default=None if _settings.device == "auto" else _settings.device)
    p.add_argument("--tokenizer", default=None,
                   help="local HF tokenizers JSON (default: synthetic)")
    p.add_argument("--weights", default=None,
                   help="HF-layout safetensors dir (default: random-init)")
    args = p.parse_args()

    def factory() -> LLM:
        import torch

        from dts_amd.serving import LocalBackend, ServingEngine

        device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
        engine = ServingEngine(
            model_name=args.model,
            device=device,
            tokenizer_path=args.tokenizer,
            weights_path=args.weights,
        )
        backend = LocalBackend.single(engine, name=args.model)
        return LLM(backend, default_model=args.model)

    app = create_app(factory)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
