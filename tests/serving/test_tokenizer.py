"""Synthetic tokenizer + chat-template tests."""

import pytest

from dts_amd.llm.types import Message
from dts_amd.serving.tokenizer import ChatTemplate, SyntheticTokenizer


@pytest.fixture
def tok():
    return SyntheticTokenizer(128256)


class TestTokenizer:
    def test_byte_roundtrip(self, tok):
        text = "Hello, wörld! 你好"
        assert tok.decode(tok.encode(text)) == text

    def test_synthetic_ids_decode_to_one_char(self, tok):
        """Sampled ids outside the byte range must decode to exactly one
        printable char — the re-encoded prompt then costs ~1 token per
        sampled token (regression test for the 6x prompt inflation)."""
        ids = [300, 50000, 128255]
        text = tok.decode(ids)
        assert len(text) == len(ids)
        assert all(32 <= ord(c) < 127 for c in text)
        assert len(tok.encode(text)) == len(ids)

    def test_specials_render_empty(self, tok):
        assert tok.decode([tok.bos_id, tok.eos_id, tok.eot_id]) == ""

    def test_too_small_vocab_rejected(self):
        with pytest.raises(ValueError):
            SyntheticTokenizer(100)


class TestChatTemplate:
    def test_render_structure(self, tok):
        tpl = ChatTemplate(tok)
        msgs = [Message.system("sys"), Message.user("hi")]
        ids = tpl.render(msgs)
        assert ids[0] == tok.bos_id
        assert ids[1] == tok.bot_id and ids[2] == tok.role_id("system")
        # ends with generation prompt
        assert ids[-2] == tok.bot_id and ids[-1] == tok.role_id("assistant")

    def test_prefix_stability_across_turns(self, tok):
        """Turn t's rendering must be a strict prefix of turn t+1's (minus
        the generation prompt) — the property block-aligned KV reuse
        depends on."""
        tpl = ChatTemplate(tok)
        msgs = [Message.system("s"), Message.user("a")]
        r1 = tpl.render(msgs, add_generation_prompt=False)
        msgs2 = msgs + [Message.assistant("b"), Message.user("c")]
        r2 = tpl.render(msgs2, add_generation_prompt=False)
        assert r2[: len(r1)] == r1

    def test_stop_tokens(self, tok):
        tpl = ChatTemplate(tok)
        assert tok.eot_id in tpl.stop_token_ids


def test_unicode_roundtrip():
    """UTF-8 byte-level: multibyte scripts and emoji round-trip exactly."""
    from dts_amd.serving.tokenizer import SyntheticTokenizer

    tok = SyntheticTokenizer(128256)
    for text in ["héllo wörld", "日本語テスト", "emoji 🎉 test", ""]:
        assert tok.decode(tok.encode(text)) == text
