"""Batched sampling front-end.

Builds per-step temperature/top-p tensors for the sampled rows, applies
constrained-decoding masks (serving/structured.py guides), and dispatches
to the top-p sampler (HIP kernel on GPU — sort-free threshold sampler —
or the torch reference on CPU). Seeded per-sequence for reproducibility.
"""

from __future__ import annotations

from typing import Optional

import torch

from dts_amd import ops


class Sampler:
    def __init__(self, device: str = "cpu") -> None:
        self.device = device
        self._generators: dict = {}  # seq_id -> torch.Generator

    def _generator_for(self, seq) -> Optional[torch.Generator]:
        if seq.params.seed is None:
            return None
        g = self._generators.get(seq.seq_id)
        if g is None:
            g = torch.Generator(device="cpu")
            g.manual_seed(int(seq.params.seed) & 0x7FFFFFFF)
            self._generators[seq.seq_id] = g
        return g

    def release(self, seq) -> None:
        self._generators.pop(seq.seq_id, None)

    def sample(self, logits: torch.Tensor, seqs: list) -> list:
        """logits [S, V] fp32 for the sampled rows; returns token ids."""
        S, V = logits.shape
        assert S == len(seqs)

        # constrained-decoding masks
        for i, seq in enumerate(seqs):
            guide = seq.guide
            if guide is not None:
                allowed = guide.allowed_tokens()
                if allowed is not None:
                    mask = torch.full((V,), float("-inf"), device=logits.device)
                    idx = torch.as_tensor(
                        allowed, dtype=torch.long, device=logits.device
                    )
                    mask[idx] = 0.0
                    logits[i] = logits[i] + mask

        temps = torch.tensor(
            [s.params.temperature for s in seqs], dtype=torch.float32
        )
        top_ps = torch.tensor([s.params.top_p for s in seqs], dtype=torch.float32)

        if logits.is_cuda:
            import random as _random

            seeds = torch.tensor(
                [
                    (hash((s.params.seed or 0, s.seq_id, len(s.tokens))) & 0x7FFFFFFF)
                    if s.params.seed is not None
                    else _random.getrandbits(31)
                    for s in seqs
                ],
                dtype=torch.long,
                device=logits.device,
            )
            toks = ops.top_p_sample(
                logits, temps.to(logits.device), top_ps.to(logits.device), seeds=seeds
            )
            return [int(t) for t in toks.cpu()]

        gens = [self._generator_for(s) for s in seqs]
        toks = ops.top_p_sample(logits.cpu(), temps, top_ps, generators=gens)
        return [int(t) for t in toks]
