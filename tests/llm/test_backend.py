"""LLM front-end tests: JSON extraction, think-strip, parse retry.

Parity coverage of ref tests/llm/test_client.py's JSON-retry and extraction
paths (ref client.py:141-203, 453-478), against a scripted backend instead
of a mocked OpenAI SDK.
"""

import pytest

from dts_amd.llm import LLM, JSONParseError, Message, ScriptedBackend
from dts_amd.llm.backend import extract_json_object, parse_json_completion, strip_think_tags


class TestExtraction:
    def test_raw_object(self):
        assert extract_json_object('{"a": 1}') == '{"a": 1}'

    def test_fenced(self):
        assert extract_json_object('```json\n{"a": 1}\n```') == '{"a": 1}'

    def test_embedded_in_prose(self):
        text = 'Sure! Here you go: {"a": {"b": [1, 2]}} hope that helps'
        assert extract_json_object(text) == '{"a": {"b": [1, 2]}}'

    def test_braces_inside_strings(self):
        text = 'x {"a": "}{", "b": 2} y'
        assert parse_json_completion(text) == {"a": "}{", "b": 2}

    def test_strip_think(self):
        assert strip_think_tags("<think>hmm\nstuff</think>answer") == "answer"

    def test_parse_failure_raises(self):
        with pytest.raises(JSONParseError):
            parse_json_completion("no json here at all")


class TestCompleteJSON:
    def test_retries_until_valid(self, run_async):
        backend = ScriptedBackend(["garbage", "also bad", '{"ok": true}'])
        llm = LLM(backend, default_model="m")
        completion = run_async(
            llm.complete([Message.user("hi")], structured_output=True)
        )
        assert completion.data == {"ok": True}
        assert len(backend.calls) == 3

    def test_exhausts_retries(self, run_async):
        backend = ScriptedBackend(["bad"] * 3)
        llm = LLM(backend, default_model="m", max_json_retries=3)
        with pytest.raises(JSONParseError):
            run_async(llm.complete([Message.user("hi")], structured_output=True))

    def test_plain_completion_strips_think(self, run_async):
        backend = ScriptedBackend(["<think>x</think>hello"])
        llm = LLM(backend, default_model="m")
        completion = run_async(llm.complete([Message.user("hi")]))
        assert completion.message.content == "hello"
