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
        # allowed-list -> padded device index tensor (guided rows)
        self._allowed_dev_cache: dict = {}

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

    # (the round-1 CPU-side gather of guided sub-logits was replaced by
    # the on-device pick below)

    def _allowed_device_tensor(
        self, allowed: list, n_max: int, device
    ) -> torch.Tensor:
        """Cached padded device tensor of an allowed-token list (form
        segments repeat their charsets every step — the H2D upload
        happens once per distinct list, not per step)."""
        key = (tuple(allowed), n_max)
        t = self._allowed_dev_cache.get(key)
        if t is None:
            if len(self._allowed_dev_cache) > 512:
                self._allowed_dev_cache.clear()
            t = torch.full((n_max,), -1, dtype=torch.long)
            t[: len(allowed)] = torch.as_tensor(allowed, dtype=torch.long)
            t = t.to(device)
            self._allowed_dev_cache[key] = t
        return t

    def _constrained_gpu(
        self, logits: torch.Tensor, guided: list, positions: list
    ) -> torch.Tensor:
        """On-device constrained pick (VERDICT round-1 #9): gather the
        sub-vocab rows into a padded [G, n_max] matrix (-1e30 padding)
        and run the SAME top-p kernel the free rows use — the host gets
        back one index per row instead of the whole sub-logit matrix."""
        import random as _random

        dev = logits.device
        n_max = max(len(a) for _, _, a in guided)
        idx = torch.stack(
            [
                self._allowed_device_tensor(allowed, n_max, dev)
                for _, _, allowed in guided
            ]
        )
        rows = torch.as_tensor(
            [i for i, _, _ in guided], dtype=torch.long, device=dev
        )
        sub = logits[rows.unsqueeze(1), idx.clamp(min=0)].float()
        sub = torch.where(idx >= 0, sub, torch.full_like(sub, -1e30))
        temps = torch.tensor(
            [s.params.temperature for _, s, _ in guided], dtype=torch.float32
        ).to(dev, non_blocking=True)
        tops = torch.tensor(
            [s.params.top_p for _, s, _ in guided], dtype=torch.float32
        ).to(dev, non_blocking=True)
        seeds = torch.tensor(
            [
                ops.mix_seed(int(s.params.seed), positions[i])
                if s.params.seed is not None
                else _random.getrandbits(31)
                for i, s, _ in guided
            ],
            dtype=torch.long,
        ).to(dev, non_blocking=True)
        return ops.top_p_sample(sub.contiguous(), temps, tops, seeds=seeds)

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
            if logits.is_cuda:
                picks = self._constrained_gpu(logits, guided, positions).cpu()
                for (i, _seq, allowed), p in zip(guided, picks.tolist()):
                    out[i] = int(allowed[p])
                return out
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
            # enqueue the free-row kernel FIRST (async), then the guided
            # sub-vocab pick kernel; one .cpu() syncs both and the guided
            # D2H is one index per row, not the sub-logit matrix
            toks_gpu = ops.top_p_sample(
                free_logits.contiguous(),
                temps,
                top_ps,
                seeds=seeds,
            )
            picks_gpu = (
                self._constrained_gpu(logits, guided, positions) if guided else None
            )
            toks = toks_gpu.cpu()
            if picks_gpu is not None:
                for (i, _seq, allowed), p in zip(guided, picks_gpu.cpu().tolist()):
                    out[i] = int(allowed[p])
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
