"""Headless runner test (parity C21: reference main.py:40-61).

Runs the repo-root `main.py` example end-to-end on the CPU tiny model with
a minimal search, and checks that the tree-state JSON checkpoint lands on
disk in the exploration-dict schema.
"""

import asyncio
import json

import main as headless


class TestHeadlessMain:
    def test_example_runs_and_saves_checkpoint(self, tmp_path, monkeypatch):
        monkeypatch.chdir(tmp_path)  # research cache writes to CWD
        out = tmp_path / "dts_output.json"
        best = asyncio.run(
            headless.run_dts_example(
                init_branches=2,
                turns_per_branch=1,
                user_intents_per_branch=1,
                rounds=1,
                deep_research=False,
                output_path=str(out),
            )
        )
        assert out.exists()
        d = json.loads(out.read_text())
        assert d["summary"]["best_score"] == best
        assert len(d["branches"]) >= 2
        for b in d["branches"]:
            assert {"id", "status", "trajectory"} <= set(b)
