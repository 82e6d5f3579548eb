"""Chained device-resident decode: many graph replays, one host sync.

The round-1 profile showed decode steps at ~7.3 ms wall against ~4 ms of
kernel time: the gap is the per-step host round-trip (sampler sync +
padded H2D rebuild + Python bookkeeping). This module removes it the
CDNA4-idiomatic way: during decode-only stretches the engine enqueues
graph replays BACK TO BACK — the sampled tokens feed the next replay's
static token input on device, positions/kv_lens/slots advance with tiny
device ops, per-step tokens stream to pinned host memory behind events —
and the host processes results lagging the GPU, syncing once per chain.

Seeding stays bit-identical to the per-step path: seeds derive on device
as mix_seed(base, position+1) (ops/hip/sampling.hip derive_seeds), the
same stateless mix the host sampler uses, so a seeded request produces
the same tokens whether it was chained, speculative, or stepped.

Correctness notes:
  - a sequence that samples its stop token (or exhausts max_tokens)
    mid-chain keeps decoding garbage rows until the host notices; those
    tokens are simply never appended, and its _finish is DEFERRED to
    chain exit so its blocks cannot be re-allocated while in-flight
    replays still write KV into them.
  - positions of real rows are bounded by reserve (max_position guard);
    pad rows restart from 0 at every chain entry.
  - new arrivals (scheduler.waiting_count() > 0) break the chain at the
    next burst boundary, so prefill work is delayed by at most
    BURST * step-time.
"""

from __future__ import annotations

import torch

from dts_amd import ops
from dts_amd.serving.batch import ForwardBatch

CHAIN_MAX = 64  # max replays per chain (also the pinned ring size)
# pipeline depth: the host keeps at most this many steps enqueued ahead
# of the last processed one. Deep enough to never starve the GPU (host
# work per step is ~50 us vs ~5 ms of GPU), shallow enough that a new
# request arriving mid-chain waits at most DEPTH steps before the chain
# yields to the scheduler (an 8-step burst cost ~50 ms of admission
# latency per arrival and erased the chain's win on the search bench).
DEPTH = 3
MIN_CHAIN = 4  # not worth the setup below this


class ChainRunner:
    """Owns the per-bucket sampling statics and the pinned token ring;
    the engine drives the loop (it holds the scheduler lock)."""

    def __init__(self, graph_runner, device: str) -> None:
        self.gr = graph_runner
        self.device = device
        self._sampling: dict = {}  # bucket -> static sampling tensors
        self._events: list = [torch.cuda.Event() for _ in range(CHAIN_MAX)]
        self._pinned: dict = {}  # bucket -> pinned [CHAIN_MAX, B] long

    # ------------------------------------------------------------------
    def eligible(self, batch: ForwardBatch) -> bool:
        seqs = getattr(batch, "_sampled_seqs", None)
        return (
            batch.num_prefill_seqs == 0
            and seqs is not None
            and len(seqs) == batch.num_decode_seqs  # no draft rows
            and not getattr(batch, "_spec_drafts", None)
            and all(s.guide is None for s in seqs)
            and self.gr.can_run(batch)
        )

    # ------------------------------------------------------------------
    def _sampling_statics(self, B: int) -> dict:
        s = self._sampling.get(B)
        if s is None:
            dev = self.device
            s = {
                "temps": torch.zeros(B, dtype=torch.float32, device=dev),
                "top_ps": torch.ones(B, dtype=torch.float32, device=dev),
                "bases": torch.zeros(B, dtype=torch.long, device=dev),
                "seeds": torch.zeros(B, dtype=torch.long, device=dev),
                "toks": torch.zeros(B, dtype=torch.long, device=dev),
                "blk": torch.zeros(B, dtype=torch.long, device=dev),
            }
            self._sampling[B] = s
        if B not in self._pinned:
            self._pinned[B] = torch.zeros(
                CHAIN_MAX, B, dtype=torch.long, pin_memory=True
            )
        return s

    def prepare(self, batch: ForwardBatch, seqs: list) -> dict:
        """Load graph statics from the batch and build sampling statics.
        Returns the chain context dict."""
        import random as _random

        entry = self.gr.entry_for(batch.num_decode_seqs)
        self.gr.load_batch(entry, batch)
        B = entry["static"]["token_ids"].shape[0]
        n = len(seqs)
        sp = self._sampling_statics(B)
        temps = torch.zeros(B, dtype=torch.float32)
        tops = torch.ones(B, dtype=torch.float32)
        bases = torch.zeros(B, dtype=torch.long)
        for i, s in enumerate(seqs):
            temps[i] = s.params.temperature
            tops[i] = s.params.top_p
            bases[i] = (
                int(s.params.seed)
                if s.params.seed is not None
                else _random.getrandbits(31)
            )
        sp["temps"].copy_(temps, non_blocking=True)
        sp["top_ps"].copy_(tops, non_blocking=True)
        sp["bases"].copy_(bases, non_blocking=True)
        return {
            "entry": entry,
            "sp": sp,
            "B": B,
            "n": n,
            "pinned": self._pinned[B],
            "block_size": self.gr.kv_pool.block_size,
        }

    # ------------------------------------------------------------------
    def launch_step(self, ctx: dict, i: int) -> None:
        """Enqueue one full chained step (no host sync anywhere)."""
        entry, sp = ctx["entry"], ctx["sp"]
        st = entry["static"]
        bs = ctx["block_size"]
        entry["graph"].replay()
        # seeds from CURRENT positions (drawn index = qpos + 1)
        ops.derive_seeds(sp["seeds"], sp["bases"], st["positions"])
        ops.top_p_sample(
            entry["logits"],
            sp["temps"],
            sp["top_ps"],
            seeds=sp["seeds"],
            out=sp["toks"],
        )
        # stream this step's tokens to pinned host memory + event
        ctx["pinned"][i].copy_(sp["toks"], non_blocking=True)
        self._events[i].record()
        # feed back: next replay decodes the sampled tokens one position on
        st["token_ids"].copy_(sp["toks"])
        st["positions"].add_(1)
        st["kv_lens"].add_(1)
        torch.div(st["positions"], bs, rounding_mode="floor", out=sp["blk"])
        tbl = torch.gather(st["block_tables"], 1, sp["blk"].view(-1, 1))
        st["slot_mapping"].copy_(
            tbl.view(-1).to(torch.long) * bs + st["positions"] % bs
        )

    def step_ready(self, i: int) -> bool:
        return self._events[i].query()

    def wait_step(self, i: int) -> None:
        self._events[i].synchronize()

    def tokens_of(self, ctx: dict, i: int) -> list:
        return ctx["pinned"][i, : ctx["n"]].tolist()
