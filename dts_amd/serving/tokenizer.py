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


class HFTokenizer:
    """Adapter over a local `tokenizers` JSON file with the same duck-type
    surface the engine uses (encode -> list[int], decode -> str, special
    ids, role headers). Special tokens are probed by name so both
    Llama-3-style (<|begin_of_text|>/<|eot_id|>) and classic BPE
    (<s>/</s>) vocabularies work."""

    def __init__(self, path: str) -> None:
        from tokenizers import Tokenizer  # local wheel, no network

        self.tk = Tokenizer.from_file(path)
        self.vocab_size = self.tk.get_vocab_size()

        def probe(*names):
            for n in names:
                i = self.tk.token_to_id(n)
                if i is not None:
                    return i
            return None

        self.bos_id = probe("<|begin_of_text|>", "<s>", "<bos>")
        self.eos_id = probe("<|end_of_text|>", "</s>", "<eos>")
        self.eot_id = probe("<|eot_id|>", "<|im_end|>")
        if self.eot_id is None:
            self.eot_id = self.eos_id
        self._hdr_start = probe("<|start_header_id|>", "<|im_start|>")
        self._hdr_end = probe("<|end_header_id|>")
        pad = probe("<pad>", "<|pad|>")
        self.pad_id = pad if pad is not None else (self.eos_id or 0)

    def encode(self, text: str, add_bos: bool = False) -> list:
        ids = self.tk.encode(text, add_special_tokens=False).ids
        if add_bos and self.bos_id is not None:
            return [self.bos_id] + ids
        return ids

    def decode(self, ids) -> str:
        return self.tk.decode(list(ids), skip_special_tokens=True)

    @property
    def byte_token_map(self) -> dict:
        """byte value -> single token id, for every printable byte the
        vocab can express as ONE token. Constrained-JSON guides
        (serving/structured.py) translate their byte-level masks through
        this; a form whose charset is not covered raises loudly."""
        if not hasattr(self, "_byte_map"):
            m = {}
            for b in range(32, 127):
                ids = self.tk.encode(chr(b), add_special_tokens=False).ids
                if len(ids) == 1:
                    m[b] = ids[0]
            self._byte_map = m
        return self._byte_map


class HFChatTemplate:
    """Chat layout over an HFTokenizer: the Llama-3 header layout when the
    header specials exist, else a plain-text role header. Token-exact so
    block-aligned prefix reuse stays deterministic."""

    def __init__(self, tok: HFTokenizer) -> None:
        self.tok = tok

    def _header(self, role: str) -> list:
        t = self.tok
        if t._hdr_start is not None and t._hdr_end is not None:
            return [t._hdr_start] + t.encode(role) + [t._hdr_end] + t.encode("\n\n")
        return t.encode(f"<{role}>\n")

    def render(self, messages: list, add_generation_prompt: bool = True) -> list:
        t = self.tok
        ids = [t.bos_id] if t.bos_id is not None else []
        for m in messages:
            ids.extend(self._header(m.role))
            ids.extend(t.encode(m.content or ""))
            if t.eot_id is not None:
                ids.append(t.eot_id)
        if add_generation_prompt:
            ids.extend(self._header("assistant"))
        return ids

    @property
    def stop_token_ids(self) -> list:
        out = [i for i in (self.tok.eot_id, self.tok.eos_id) if i is not None]
        return out or [0]


def load_tokenizer(spec_vocab_size: int, path: Optional[str] = None):
    """Return an HFTokenizer when a local JSON file is given, else the
    synthetic byte-level tokenizer at the spec's vocab width."""
    if path:
        return HFTokenizer(path)
    return SyntheticTokenizer(spec_vocab_size)
