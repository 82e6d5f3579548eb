"""Prompt-lookup (n-gram) draft proposal for speculative decoding.

The search workload constantly re-decodes text it has already seen —
user/assistant turns quote and paraphrase the conversation history, judge
critiques echo trajectory phrasing — so an n-gram lookup over the
sequence's OWN tokens (prompt + generated) is a free draft model: no
second network, no extra weights streamed.

Verification is exact-match-by-sampling: every draft row is an
independent decode row (own position / kv_len); the sampler draws the
target token at each position with a stateless per-(seed, position) RNG,
and the engine emits sampled[j] while sampled[j-1] == draft[j-1]. Each
emitted token is the target model's OWN sample conditioned on accepted
history, so the output distribution is exactly the non-speculative one —
acceptance only decides how many samples land per step.

The native scheduler (dts_amd/core/csrc/core.cpp) carries the same
bigram index in C++; this Python twin defines the reference semantics and
the differential tests pin the two together.
"""

from __future__ import annotations


class NgramIndex:
    """Incremental bigram → continuation-position index over one sequence.

    For bigram (a, b) ending at token index i, the continuation position
    is i + 1 (the index of the token that followed it). We keep the two
    most recent continuations: at propose time the latest one is always
    the sequence's own tail (self-match), so the previous one is the
    draft source.
    """

    __slots__ = ("_map", "_n")

    def __init__(self, tokens: list | None = None) -> None:
        self._map: dict = {}  # (a, b) -> [latest_cont, prev_cont]
        self._n = 0
        if tokens:
            self.extend(tokens)

    def extend(self, tokens: list) -> None:
        """Index new tokens; `tokens` is the FULL live list (appended-to)."""
        L = len(tokens)
        for i in range(max(1, self._n), L):
            key = (tokens[i - 1], tokens[i])
            ent = self._map.get(key)
            if ent is None:
                self._map[key] = [i + 1, 0]
            else:
                ent[1] = ent[0]
                ent[0] = i + 1
        self._n = L

    def propose(self, tokens: list, max_k: int) -> list:
        """Draft tokens predicted to follow tokens[-1]."""
        L = len(tokens)
        if L < 3 or max_k <= 0:
            return []
        ent = self._map.get((tokens[-2], tokens[-1]))
        if ent is None:
            return []
        cont = ent[1] if ent[0] >= L else ent[0]
        if cont <= 0 or cont >= L:
            return []
        k = min(max_k, L - cont)
        return list(tokens[cont : cont + k])
