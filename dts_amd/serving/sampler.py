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

    def _sample_constrained(self, row: torch.Tensor, seq, allowed: list) -> int:
        """Sample over a SMALL allowed set (constrained decoding): gather
        the allowed logits instead of masking the full 128k vocab — the
        set is typically 10-95 bytes, so this is a tiny CPU softmax."""
        idx = torch.as_tensor(allowed, dtype=torch.long, device=row.device)
        sub = row[idx].float().cpu()
        t = seq.params.temperature
        if t <= 0.0:
            return int(idx[int(sub.argmax())])
        probs = torch.softmax(sub / t, dim=-1)
        p = seq.params.top_p
        if p < 1.0 and probs.numel() > 1:
            sp, si = torch.sort(probs, descending=True)
            cum = torch.cumsum(sp, -1)
            keep = (cum - sp) < p
            keep[0] = True
            kept = sp * keep
            kept = kept / kept.sum()
            pick = torch.multinomial(kept, 1, generator=self._generator_for(seq))
            return int(idx[int(si[pick])])
        pick = torch.multinomial(probs, 1, generator=self._generator_for(seq))
        return int(idx[pick])

    def sample(self, logits: torch.Tensor, seqs: list) -> list:
        """logits [S, V] fp32 for the sampled rows; returns token ids."""
        S, V = logits.shape
        assert S == len(seqs)

        out: list = [None] * S
        free_rows: list = []
        guided: list = []  # (i, seq, allowed)
        for i, seq in enumerate(seqs):
            allowed = seq.guide.allowed_tokens() if seq.guide is not None else None
            if allowed is not None:
                guided.append((i, seq, allowed))
            else:
                free_rows.append(i)

        if not free_rows:
            for i, seq, allowed in guided:
                out[i] = self._sample_constrained(logits[i], seq, allowed)
            return out

        free_seqs = [seqs[i] for i in free_rows]
        free_logits = (
            logits
            if len(free_rows) == S
            else logits[torch.as_tensor(free_rows, device=logits.device)]
        )
        temps = torch.tensor(
            [s.params.temperature for s in free_seqs], dtype=torch.float32
        )
        top_ps = torch.tensor([s.params.top_p for s in free_seqs], dtype=torch.float32)

        if logits.is_cuda:
            import random as _random

            seeds = torch.tensor(
                [
                    (hash((s.params.seed or 0, s.seq_id, len(s.tokens))) & 0x7FFFFFFF)
                    if s.params.seed is not None
                    else _random.getrandbits(31)
                    for s in free_seqs
                ],
                dtype=torch.long,
                device=logits.device,
            )
            # enqueue the free-row kernel FIRST (async), then the guided
            # gathers — their .cpu() syncs then overlap one GPU drain
            toks_gpu = ops.top_p_sample(
                free_logits.contiguous(),
                temps.to(logits.device),
                top_ps.to(logits.device),
                seeds=seeds,
            )
            for i, seq, allowed in guided:
                out[i] = self._sample_constrained(logits[i], seq, allowed)
            toks = toks_gpu.cpu()
        else:
            for i, seq, allowed in guided:
                out[i] = self._sample_constrained(logits[i], seq, allowed)
            gens = [self._generator_for(s) for s in free_seqs]
            toks = ops.top_p_sample(free_logits.cpu(), temps, top_ps, generators=gens)
        for i, t in zip(free_rows, toks):
            out[i] = int(t)
        return out
