"""Tool / registry / agentic-loop tests (parity: ref tests/llm/test_tools.py)."""

import json

import pytest

from dts_amd.llm import LLM, ScriptedBackend
from dts_amd.llm.tools import Tool, ToolRegistry, run_tool_loop


def add(a: int, b: int) -> int:
    """Add two integers."""
    return a + b


async def async_upper(text: str) -> str:
    """Uppercase text."""
    return text.upper()


class TestTool:
    def test_schema_from_signature(self):
        t = Tool(add)
        schema = t.to_schema()
        assert schema["function"]["name"] == "add"
        assert schema["function"]["description"] == "Add two integers."
        params = schema["function"]["parameters"]
        assert params["properties"]["a"]["type"] == "integer"
        assert set(params["required"]) == {"a", "b"}

    def test_execute_sync_and_async(self, run_async):
        assert run_async(Tool(add).execute({"a": 2, "b": 3})) == 5
        assert run_async(Tool(async_upper).execute({"text": "hi"})) == "HI"


class TestRegistry:
    def test_register_and_lookup(self):
        reg = ToolRegistry()
        reg.register(add)
        reg.register(async_upper, name="shout")
        assert len(reg) == 2
        assert "add" in reg and "shout" in reg
        assert reg.get("missing") is None
        assert len(reg.schemas()) == 2


class TestToolLoop:
    def test_tool_then_answer(self, run_async):
        backend = ScriptedBackend(
            [
                json.dumps({"tool": "add", "arguments": {"a": 4, "b": 5}}),
                json.dumps({"tool": None, "answer": "the sum is 9"}),
            ]
        )
        llm = LLM(backend, default_model="m")
        reg = ToolRegistry()
        reg.register(add)
        from dts_amd.llm.types import Message

        completion = run_async(
            run_tool_loop(llm, [Message.user("what is 4+5?")], reg)
        )
        assert completion.message.content == "the sum is 9"
        # tool result fed back into the conversation
        second_call = backend.calls[1]
        assert any("Tool result: 9" in (m.content or "") for m in second_call)

    def test_unknown_tool_recovers(self, run_async):
        backend = ScriptedBackend(
            [
                json.dumps({"tool": "nope", "arguments": {}}),
                json.dumps({"tool": None, "answer": "done"}),
            ]
        )
        llm = LLM(backend, default_model="m")
        reg = ToolRegistry()
        from dts_amd.llm.types import Message

        completion = run_async(run_tool_loop(llm, [Message.user("x")], reg))
        assert completion.message.content == "done"

    def test_tool_error_surfaced(self, run_async):
        def boom() -> int:
            raise ValueError("nope")

        backend = ScriptedBackend(
            [
                json.dumps({"tool": "boom", "arguments": {}}),
                json.dumps({"tool": None, "answer": "recovered"}),
            ]
        )
        llm = LLM(backend, default_model="m")
        reg = ToolRegistry()
        reg.register(boom)
        from dts_amd.llm.types import Message

        completion = run_async(run_tool_loop(llm, [Message.user("x")], reg))
        assert completion.message.content == "recovered"
        second_call = backend.calls[1]
        assert any("error" in (m.content or "") for m in second_call)
