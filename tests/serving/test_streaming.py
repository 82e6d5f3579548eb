"""Token-streaming tests on the tiny CPU engine."""

import asyncio

import pytest
import torch

from dts_amd.llm import LLM
from dts_amd.llm.types import Message, SamplingParams
from dts_amd.serving import LocalBackend, ServingEngine


@pytest.fixture(scope="module")
def backend():
    eng = ServingEngine(
        model_name="llama-tiny",
        device="cpu",
        dtype=torch.float32,
        num_blocks=512,
        block_size=8,
        weight_seed=2,
    )
    b = LocalBackend.single(eng, name="llama-tiny")
    yield b
    b.shutdown()


def test_stream_matches_complete(backend):
    msgs = [Message.system("s"), Message.user("tell me something")]

    async def main():
        llm = LLM(backend, default_model="llama-tiny")
        chunks = []
        async for delta in llm.stream(msgs, max_tokens=12, seed=5, temperature=0.7):
            chunks.append(delta)
        whole = await llm.complete(msgs, max_tokens=12, seed=5, temperature=0.7)
        return "".join(chunks), whole.message.content

    streamed, whole = asyncio.run(main())
    assert streamed  # produced something
    # deterministic engine: same seed → same text through either path
    # (complete() whitespace-strips the final text; stream deltas do not)
    assert streamed.strip() == whole


def test_stream_fallback_without_backend_support(run_async):
    from dts_amd.llm import ScriptedBackend

    llm = LLM(ScriptedBackend(["hello world"]), default_model="m")

    async def main():
        out = []
        async for d in llm.stream([Message.user("x")]):
            out.append(d)
        return out

    assert run_async(main()) == ["hello world"]
