from dts_amd.llm.types import Completion, Message, Usage
from dts_amd.llm.errors import (
    BackendError,
    ContextLengthError,
    EmptyResponseError,
    JSONParseError,
    LLMError,
    TimeoutError_,
)
from dts_amd.llm.backend import InferenceBackend, LLM
from dts_amd.llm.fake import FakeBackend, ScriptedBackend

__all__ = [
    "Completion",
    "Message",
    "Usage",
    "LLMError",
    "BackendError",
    "ContextLengthError",
    "EmptyResponseError",
    "JSONParseError",
    "TimeoutError_",
    "InferenceBackend",
    "LLM",
    "FakeBackend",
    "ScriptedBackend",
]
