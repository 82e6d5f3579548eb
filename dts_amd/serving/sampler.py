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
        # (temps, top_ps) device tensors cached per batch composition —
        # stable across decode stretches; saves two H2D uploads per step
        self._param_cache: dict = {}

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

    def _gather_constrained(self, logits: torch.Tensor, guided: list):
        """One flat gather + ONE device sync for ALL guided rows (a
        per-row .cpu() costs a stream sync each)."""
        flat_rows: list = []
        flat_idx: list = []
        for i, _seq, allowed in guided:
            flat_rows.extend([i] * len(allowed))
            flat_idx.extend(allowed)
        sub = logits[
            torch.as_tensor(flat_rows, dtype=torch.long, device=logits.device),
            torch.as_tensor(flat_idx, dtype=torch.long, device=logits.device),
        ].float()
        return sub  # caller moves to CPU (the single sync point)

    def _pick_constrained(self, sub_cpu: torch.Tensor, seq, allowed: list) -> int:
        t = seq.params.temperature
        if t <= 0.0:
            return int(allowed[int(sub_cpu.argmax())])
        probs = torch.softmax(sub_cpu / t, dim=-1)
        p = seq.params.top_p
        if p < 1.0 and probs.numel() > 1:
            sp, si = torch.sort(probs, descending=True)
            cum = torch.cumsum(sp, -1)
            keep = (cum - sp) < p
            keep[0] = True
            kept = sp * keep
            kept = kept / kept.sum()
            pick = torch.multinomial(kept, 1, generator=self._generator_for(seq))
            return int(allowed[int(si[pick])])
        pick = torch.multinomial(probs, 1, generator=self._generator_for(seq))
        return int(allowed[int(pick)])

    def _sample_constrained(self, row: torch.Tensor, seq, allowed: list) -> int:
        """Single-row convenience path (CPU backend / no free rows)."""
        idx = torch.as_tensor(allowed, dtype=torch.long, device=row.device)
        sub = row[idx].float().cpu()
        return self._pick_constrained(sub, seq, allowed)

    def sample(
        self, logits: torch.Tensor, seqs: list, positions: Optional[list] = None
    ) -> list:
        """logits [S, V] fp32 for the sampled rows; returns token ids.

        positions[i] = token index row i draws (defaults to
        len(seq.tokens)); speculative draft rows pass len+j so the seeded
        stream is identical with speculation on or off — seeding is
        stateless in (seed, position) on CPU and GPU alike.
        """
        S, V = logits.shape
        assert S == len(seqs)
        if positions is None:
            positions = [len(s.tokens) for s in seqs]

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
        free_pos = [positions[i] for i in free_rows]
        free_logits = (
            logits
            if len(free_rows) == S
            else logits[torch.as_tensor(free_rows, device=logits.device)]
        )
        key = tuple((s.params.temperature, s.params.top_p) for s in free_seqs)
        cached = self._param_cache.get(key)
        if cached is None:
            temps = torch.tensor([t for t, _ in key], dtype=torch.float32)
            top_ps = torch.tensor([p for _, p in key], dtype=torch.float32)
            if logits.is_cuda:
                temps = temps.to(logits.device)
                top_ps = top_ps.to(logits.device)
            if len(self._param_cache) > 256:
                self._param_cache.clear()
            self._param_cache[key] = (temps, top_ps)
            cached = (temps, top_ps)
        temps, top_ps = cached

        if logits.is_cuda:
            import random as _random

            seeds = torch.tensor(
                [
                    # mix(seed, position) only — the stream must not
                    # depend on seq_id (submission order / co-batched
                    # traffic), stays identical across preempt+recompute
                    # or speculative re-derivation, and matches the
                    # on-device derive_seeds of the chained decode loop
                    ops.mix_seed(int(s.params.seed), p)
                    if s.params.seed is not None
                    else _random.getrandbits(31)
                    for s, p in zip(free_seqs, free_pos)
                ],
                dtype=torch.long,
                device=logits.device,
            )
            # enqueue the free-row kernel FIRST (async), then ONE flat
            # gather for all guided rows; a single .cpu() syncs both
            toks_gpu = ops.top_p_sample(
                free_logits.contiguous(),
                temps,
                top_ps,
                seeds=seeds,
            )
            if guided:
                sub_cpu = self._gather_constrained(logits, guided).cpu()
                off = 0
                for i, seq, allowed in guided:
                    n_a = len(allowed)
                    out[i] = self._pick_constrained(
                        sub_cpu[off : off + n_a], seq, allowed
                    )
                    off += n_a
            toks = toks_gpu.cpu()
        else:
            for i, seq, allowed in guided:
                out[i] = self._sample_constrained(logits[i], seq, allowed)
            # stateless per-(seed, position) generators, matching the GPU
            # kernel's seeding discipline — a persistent generator would
            # advance on unaccepted draft rows and diverge with spec on
            gens = []
            for s, p in zip(free_seqs, free_pos):
                if s.params.seed is None:
                    gens.append(None)
                else:
                    g = torch.Generator(device="cpu")
                    g.manual_seed(ops.mix_seed(int(s.params.seed), p))
                    gens.append(g)
            toks = ops.top_p_sample(free_logits.cpu(), temps, top_ps, generators=gens)
        for i, t in zip(free_rows, toks):
            out[i] = int(t)
        return out
