"""Server tests (parity coverage of ref tests/api/test_server.py).

Uses FastAPI TestClient with the deterministic FakeBackend as the
inference path — the full search runs behind the websocket."""

import json

import pytest
from fastapi.testclient import TestClient

from dts_amd.llm import LLM, FakeBackend
from dts_amd.server.app import create_app
from dts_amd.server.schemas import SearchRequest


@pytest.fixture
def client():
    app = create_app(lambda: LLM(FakeBackend(), default_model="fake"))
    with TestClient(app) as c:
        yield c


class TestRest:
    def test_health(self, client):
        r = client.get("/health")
        assert r.status_code == 200
        assert r.json() == {"status": "ok"}

    def test_config_defaults(self, client):
        r = client.get("/config")
        body = r.json()
        assert body["init_branches"] == 6
        assert body["turns_per_branch"] == 5
        assert body["scoring_mode"] == "comparative"
        assert body["prune_threshold"] == 6.5
        # additive fields present
        assert body["user_variability"] is False

    def test_models_list(self, client):
        r = client.get("/api/models")
        data = r.json()["data"]
        ids = {m["id"] for m in data}
        assert "llama-3-8b" in ids
        assert all(m["architecture"]["modality"] == "text->text" for m in data)

    def test_index(self, client):
        r = client.get("/")
        assert r.status_code == 200


class TestSchemas:
    def test_request_bounds(self):
        with pytest.raises(Exception):
            SearchRequest(goal="g", first_message="m", init_branches=0)
        with pytest.raises(Exception):
            SearchRequest(goal="g", first_message="m", prune_threshold=11)
        req = SearchRequest(goal="g", first_message="m")
        assert req.rounds == 1 and req.scoring_mode == "comparative"

    def test_additive_fields_roundtrip(self):
        req = SearchRequest(
            goal="g", first_message="m", user_variability=True, reasoning_enabled=True
        )
        assert req.user_variability and req.reasoning_enabled


class TestWebSocket:
    def test_ping_pong(self, client):
        with client.websocket_connect("/ws") as ws:
            ws.send_text(json.dumps({"type": "ping"}))
            msg = json.loads(ws.receive_text())
            assert msg["type"] == "pong"

    def test_invalid_json(self, client):
        with client.websocket_connect("/ws") as ws:
            ws.send_text("not json")
            msg = json.loads(ws.receive_text())
            assert msg["type"] == "error"

    def test_invalid_config(self, client):
        with client.websocket_connect("/ws") as ws:
            ws.send_text(
                json.dumps({"type": "start_search", "config": {"goal": "only"}})
            )
            msg = json.loads(ws.receive_text())
            assert msg["type"] == "error"
            assert "invalid config" in msg["data"]["message"]

    def test_full_search_stream(self, client):
        with client.websocket_connect("/ws") as ws:
            ws.send_text(
                json.dumps(
                    {
                        "type": "start_search",
                        "config": {
                            "goal": "teach fractions",
                            "first_message": "can you help me learn fractions?",
                            "init_branches": 2,
                            "turns_per_branch": 1,
                            "scoring_mode": "absolute",
                            "prune_threshold": 0.0,
                            "rounds": 1,
                        },
                    }
                )
            )
            from dts_amd.server import schemas as sch

            validators = {
                "search_started": sch.SearchStartedData,
                "phase": sch.PhaseData,
                "strategy_generated": sch.StrategyGeneratedData,
                "node_added": sch.NodeAddedData,
                "node_updated": sch.NodeUpdatedData,
                "round_started": sch.RoundStartedData,
                "nodes_pruned": sch.NodesPrunedData,
                "intent_generated": sch.IntentGeneratedData,
                "token_update": sch.TokenUpdateData,
                "error": sch.ErrorData,
            }
            events = []
            while True:
                msg = json.loads(ws.receive_text())
                events.append(msg["type"])
                model = validators.get(msg["type"])
                if model is not None:
                    model(**msg["data"])  # event data matches the wire schema
                if msg["type"] in ("complete", "error"):
                    final = msg
                    break
            assert final["type"] == "complete"
            sch.CompleteData(**final["data"])
            assert "search_started" in events
            assert "node_added" in events
            assert "node_updated" in events
            exploration = final["data"]["exploration"]
            assert exploration["summary"]["total_branches"] == 2
            assert {"summary", "best_branch", "branches", "research_report"} <= set(
                exploration
            )


def test_metrics_endpoint():
    """Prometheus /metrics exposes the live engine counters."""
    import torch
    from fastapi.testclient import TestClient

    from dts_amd.llm import LLM
    from dts_amd.server.app import create_app
    from dts_amd.serving import LocalBackend, ServingEngine

    eng = ServingEngine(
        model_name="llama-tiny",
        device="cpu",
        dtype=torch.float32,
        num_blocks=64,
        block_size=8,
        weight_seed=1,
    )
    backend = LocalBackend.single(eng, name="llama-tiny")
    app = create_app(llm_factory=lambda: LLM(backend, default_model="llama-tiny"))
    client = TestClient(app)
    # force backend creation (metrics reads app.state.llm lazily)
    with client.websocket_connect("/ws") as ws:
        ws.send_text('{"type": "ping"}')
        assert ws.receive_json()["type"] == "pong"
    app.state.llm = LLM(backend, default_model="llama-tiny")
    r = client.get("/metrics")
    assert r.status_code == 200
    assert 'dts_engine_stat{model="llama-tiny",stat="free_blocks"}' in r.text
    assert 'stat="preemptions"' in r.text
    backend.shutdown()


def test_frontend_js_syntax():
    """The bundled visualizer's script must be valid JavaScript."""
    import shutil
    import subprocess
    from pathlib import Path

    node = shutil.which("node")
    if node is None:
        import pytest

        pytest.skip("node not available")
    html = (
        Path(__file__).resolve().parents[2]
        / "dts_amd"
        / "server"
        / "static"
        / "index.html"
    ).read_text()
    script = html.split("<script>")[1].split("</script>")[0]
    import tempfile

    with tempfile.NamedTemporaryFile("w", suffix=".js", delete=False) as f:
        f.write(script)
        path = f.name
    p = subprocess.run([node, "--check", path], capture_output=True, text=True)
    assert p.returncode == 0, p.stderr


def test_frontend_event_handling_headless():
    """Drive the visualizer's WS event handlers in node with a stub DOM:
    node_added/node_updated/nodes_pruned must build the expected tree."""
    import shutil
    import subprocess
    from pathlib import Path

    node = shutil.which("node")
    if node is None:
        import pytest

        pytest.skip("node not available")
    root = Path(__file__).resolve().parents[2]
    p = subprocess.run(
        [
            node,
            str(Path(__file__).parent / "fe_harness.js"),
            str(root / "dts_amd" / "server" / "static" / "index.html"),
        ],
        capture_output=True,
        text=True,
    )
    assert p.returncode == 0, p.stderr
    assert "frontend logic OK" in p.stdout
