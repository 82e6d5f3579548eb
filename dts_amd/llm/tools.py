"""Function-calling support.

Parity: reference backend/llm/tools.py:20-256 — `Tool` wraps a Python
function into an OpenAI-style schema derived from its signature and
docstring; `ToolRegistry` holds a set of tools and dispatches execution
(sync or async). The reference notes this is unused by the DTS engine
 — it is part of the client library surface. Locally,
tool-call emission uses the structured-output path: `LLM.run` asks the
model for a JSON {"tool": ..., "arguments": {...}} object per step.
"""

from __future__ import annotations

import asyncio
import inspect
import json
from typing import Any, Callable, Optional

from dts_amd.llm.backend import LLM
from dts_amd.llm.types import Completion, Message

_PY_TO_JSON = {
    int: "integer",
    float: "number",
    str: "string",
    bool: "boolean",
    list: "array",
    dict: "object",
}


class Tool:
    """Wraps a callable into a named tool with a JSON-schema signature."""

    def __init__(
        self,
        fn: Callable,
        name: Optional[str] = None,
        description: Optional[str] = None,
    ) -> None:
        self.fn = fn
        self.name = name or fn.__name__
        self.description = description or (inspect.getdoc(fn) or "").split("\n")[0]
        self.parameters = self._schema_from_signature(fn)

    @staticmethod
    def _schema_from_signature(fn: Callable) -> dict:
        sig = inspect.signature(fn)
        props: dict = {}
        required: list = []
        for pname, param in sig.parameters.items():
            if pname in ("self", "cls"):
                continue
            ann = param.annotation
            jtype = _PY_TO_JSON.get(ann, "string")
            props[pname] = {"type": jtype}
            if param.default is inspect.Parameter.empty:
                required.append(pname)
        return {"type": "object", "properties": props, "required": required}

    def to_schema(self) -> dict:
        return {
            "type": "function",
            "function": {
                "name": self.name,
                "description": self.description,
                "parameters": self.parameters,
            },
        }

    async def execute(self, arguments: dict) -> Any:
        if inspect.iscoroutinefunction(self.fn):
            return await self.fn(**arguments)
        return await asyncio.get_event_loop().run_in_executor(
            None, lambda: self.fn(**arguments)
        )


class ToolRegistry:
    def __init__(self) -> None:
        self._tools: dict = {}

    def register(self, fn_or_tool, **kw) -> Tool:
        tool = fn_or_tool if isinstance(fn_or_tool, Tool) else Tool(fn_or_tool, **kw)
        self._tools[tool.name] = tool
        return tool

    def get(self, name: str) -> Optional[Tool]:
        return self._tools.get(name)

    def schemas(self) -> list:
        return [t.to_schema() for t in self._tools.values()]

    def __len__(self) -> int:
        return len(self._tools)

    def __contains__(self, name: str) -> bool:
        return name in self._tools


async def run_tool_loop(
    llm: LLM,
    messages: list,
    registry: ToolRegistry,
    model: Optional[str] = None,
    max_steps: int = 8,
) -> Completion:
    """Agentic loop (parity with ref client.py:274-330 `LLM.run`): ask the
    model to either answer or emit a {"tool": name, "arguments": {...}}
    JSON object; execute and feed results back until a final answer."""
    tool_doc = json.dumps(registry.schemas(), indent=1)
    convo = list(messages)
    convo.insert(
        0,
        Message.system(
            "You may call tools. To call one, reply with ONLY a JSON object "
            '{"tool": "<name>", "arguments": {...}}. Available tools:\n'
            + tool_doc
            + '\nWhen done, reply with {"tool": null, "answer": "<final answer>"}.'
        ),
    )
    for _ in range(max_steps):
        completion = await llm.complete(
            convo, model=model, structured_output=True
        )
        data = completion.data or {}
        tool_name = data.get("tool")
        if not tool_name:
            completion.message.content = data.get(
                "answer", completion.message.content
            )
            return completion
        tool = registry.get(tool_name)
        if tool is None:
            convo.append(Message.assistant(json.dumps(data)))
            convo.append(Message.user(f"Tool '{tool_name}' does not exist."))
            continue
        try:
            result = await tool.execute(data.get("arguments", {}) or {})
            payload = json.dumps(result, default=str)
        except Exception as e:  # noqa: BLE001 — surfaced to the model
            payload = json.dumps({"error": str(e)})
        convo.append(Message.assistant(json.dumps(data)))
        convo.append(Message.user(f"Tool result: {payload}"))
    return completion
