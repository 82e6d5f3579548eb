"""Real (HF `tokenizers`) tokenizer through the serving engine.

Trains a tiny byte-level BPE OFFLINE (no network), then runs the engine
end-to-end with it: plain generation, the chat template, and
constrained-JSON guided decoding through the byte→token translation
layer (SURVEY.md §2.3 row "Tokenizer (HF tokenizers or own BPE)").
"""

import json

import pytest
import torch

from dts_amd.llm.types import Message, SamplingParams
from dts_amd.serving import ServingEngine
from dts_amd.serving.tokenizer import HFChatTemplate, HFTokenizer


@pytest.fixture(scope="module")
def bpe_path(tmp_path_factory):
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=480,
        special_tokens=[
            "<|begin_of_text|>",
            "<|end_of_text|>",
            "<|eot_id|>",
            "<|start_header_id|>",
            "<|end_header_id|>",
        ],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    corpus = [
        "hello world, can you help me decide between two databases? " * 4,
        '{"goal": "text", "nodes": {"Strategy 1: name": "description"}}',
    ]
    tok.train_from_iterator(corpus, trainer)
    path = tmp_path_factory.mktemp("tok") / "tiny_bpe.json"
    tok.save(str(path))
    return str(path)


class TestHFTokenizer:
    def test_adapter_roundtrip_and_specials(self, bpe_path):
        t = HFTokenizer(bpe_path)
        assert t.bos_id is not None and t.eot_id is not None
        text = "hello world, a test"
        assert t.decode(t.encode(text)) == text
        # every printable ASCII byte must be single-token (ByteLevel
        # alphabet guarantees it) so guides can translate their masks
        assert set(range(32, 127)) <= set(t.byte_token_map)

    def test_chat_template_stops_and_headers(self, bpe_path):
        t = HFTokenizer(bpe_path)
        tpl = HFChatTemplate(t)
        ids = tpl.render([Message.system("s"), Message.user("u")])
        assert ids[0] == t.bos_id
        assert ids.count(t.eot_id) == 2
        assert t.eot_id in tpl.stop_token_ids

    def test_engine_generates_with_hf_tokenizer(self, bpe_path):
        eng = ServingEngine(
            model_name="llama-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=512,
            block_size=8,
            weight_seed=1,
            tokenizer_path=bpe_path,
        )
        assert isinstance(eng.tokenizer, HFTokenizer)
        ids = eng.template.render([Message.user("hello, which database?")])
        fut = eng.submit_tokens(ids, SamplingParams(max_tokens=8, temperature=0.0))
        eng.run_until_idle()
        res = fut.result(timeout=10)
        assert res.completion_tokens >= 1
        assert isinstance(res.text, str)
        eng.stop()

    def test_guided_json_through_byte_translation(self, bpe_path):
        from dts_amd.serving.structured import strategy_form

        eng = ServingEngine(
            model_name="llama-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=2048,
            block_size=8,
            weight_seed=1,
            tokenizer_path=bpe_path,
        )
        g = strategy_form(eng.tokenizer, 2)
        ids = eng.template.render([Message.user("plan strategies")])
        fut = eng.submit_tokens(
            ids, SamplingParams(max_tokens=4096, temperature=0.0), guide=g
        )
        eng.run_until_idle()
        res = fut.result(timeout=10)
        assert res.finish_reason == "stop"
        d = json.loads(res.text)
        assert len(d["nodes"]) == 2
        eng.stop()


def test_plain_bpe_without_chat_specials(tmp_path):
    """A vocab with only <s>/</s> (no Llama-3 headers) uses the
    plain-text role-header fallback and still generates."""
    import torch
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    from dts_amd.llm.types import Message, SamplingParams
    from dts_amd.serving import ServingEngine
    from dts_amd.serving.tokenizer import HFChatTemplate, HFTokenizer

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=400,
        special_tokens=["<s>", "</s>"],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    tok.train_from_iterator(["plain corpus text for a classic bpe " * 4], trainer)
    path = tmp_path / "plain.json"
    tok.save(str(path))

    t = HFTokenizer(str(path))
    assert t.bos_id is not None and t._hdr_start is None
    tpl = HFChatTemplate(t)
    ids = tpl.render([Message.user("hello")])
    assert ids[0] == t.bos_id and t.eot_id in ids  # eot falls back to </s>

    eng = ServingEngine(
        model_name="llama-tiny",
        device="cpu",
        dtype=torch.float32,
        num_blocks=256,
        block_size=8,
        weight_seed=1,
        tokenizer_path=str(path),
    )
    fut = eng.submit_tokens(ids, SamplingParams(max_tokens=6, temperature=0.0))
    eng.run_until_idle()
    assert fut.result(timeout=10).completion_tokens >= 1
    eng.stop()
