"""Tokenizers for the serving engine.

`SyntheticTokenizer` maps UTF-8 bytes into a full-size model vocabulary so
that the lm_head GEMM, softmax and sampler all run at the real model's
vocab width (e.g. Llama-3's 128256) without needing downloadable tokenizer
files — there is no network in this environment. Ids outside the byte range
(which temperature sampling over a random-init model will produce) decode
to stable pseudo-words, so round-tripping text through the model yields
valid synthetic conversation text.

`HFTokenizer` wraps a local `tokenizers` JSON file when one exists on disk.
"""

from __future__ import annotations

from typing import Optional


class SyntheticTokenizer:
    """Byte-level tokenizer padded to an arbitrary vocab size.

    Layout: [0..255] raw bytes, then special tokens, then synthetic ids.
    """

    def __init__(self, vocab_size: int) -> None:
        if vocab_size < 300:
            raise ValueError("vocab_size too small for byte tokenizer + specials")
        self.vocab_size = vocab_size
        self.bos_id = 256
        self.eos_id = 257
        self.pad_id = 258
        # chat-structure specials (used by the chat template)
        self.bot_id = 259  # begin-of-turn
        self.eot_id = 260  # end-of-turn
        self._role_ids = {"system": 261, "user": 262, "assistant": 263}
        self.n_special = 264

    def encode(self, text: str, add_bos: bool = False) -> list:
        ids = [b for b in text.encode("utf-8")]
        return ([self.bos_id] + ids) if add_bos else ids

    def decode(self, ids) -> str:
        """Ids in the byte range decode to their bytes; synthetic ids
        (what sampling over a random-init model mostly produces) decode to
        ONE printable char each, so text that re-enters a prompt re-encodes
        to ~1 token per originally sampled token — matching how a real
        BPE round-trips its own output. (A multi-char mapping here would
        silently inflate every subsequent prompt by that factor.)"""
        out = []
        byte_buf = bytearray()

        def flush():
            nonlocal byte_buf
            if byte_buf:
                out.append(byte_buf.decode("utf-8", errors="replace"))
                byte_buf = bytearray()

        for i in ids:
            i = int(i)
            if i < 256:
                byte_buf.append(i)
            elif i < self.n_special:
                flush()  # specials render as nothing
            else:
                flush()
                h = i % 95  # printable ASCII 32..126
                out.append(chr(32 + h))
        flush()
        return "".join(out)

    def role_id(self, role: str) -> int:
        return self._role_ids.get(role, self._role_ids["user"])


class ChatTemplate:
    """Token-level chat layout.

    <bot><role> content-bytes <eot> per message; generation is primed with
    <bot><assistant>. Keeping the template token-exact (not string-level)
    makes block-aligned prefix reuse across turns deterministic
    (serving/kv_cache.py).
    """

    def __init__(self, tok: SyntheticTokenizer) -> None:
        self.tok = tok

    def render(self, messages: list, add_generation_prompt: bool = True) -> list:
        ids = [self.tok.bos_id]
        for m in messages:
            ids.append(self.tok.bot_id)
            ids.append(self.tok.role_id(m.role))
            ids.extend(self.tok.encode(m.content or ""))
            ids.append(self.tok.eot_id)
        if add_generation_prompt:
            ids.append(self.tok.bot_id)
            ids.append(self.tok.role_id("assistant"))
        return ids

    @property
    def stop_token_ids(self) -> list:
        return [self.tok.eot_id, self.tok.eos_id]


def load_tokenizer(spec_vocab_size: int, path: Optional[str] = None):
    """Return an HF tokenizer when a local file is given, else synthetic."""
    if path:
        from tokenizers import Tokenizer  # local wheel, no network

        return Tokenizer.from_file(path)
    return SyntheticTokenizer(spec_vocab_size)
