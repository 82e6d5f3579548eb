"""hipGraph-captured decode steps.

The microbench (scripts/microbench.py, profiles/) showed the decode
forward flat at ~9 ms from B=1 to B=32 — launch-bound: ~350 eager kernel
launches per step against ~2.5 ms of weight-streaming math. The fix is
the CDNA4-idiomatic one (guide: "capture launch-bound inner loops in
hipGraphs"): capture one decode forward per batch-size bucket with static
input/output buffers and replay it (~10-16 µs replay floor vs ~350
launches).

Only decode-only batches replay through graphs; mixed/prefill batches run
eager (their launches amortize over thousands of tokens). Pad rows of a
bucket write their (token 0, pos 0) KV to a reserved scratch block and
attend 1 key of it — never a live block.
"""

from __future__ import annotations

import torch

from dts_amd.serving.batch import ForwardBatch
from dts_amd.utils.logging import logger

# measured (profiles/microbench_*): graphed decode at B>=24 replays 3-5x
# slower than eager (suspected capture-unfriendly hipBLASLt algo choice at
# those M) — graphs pay off at the small-B launch-bound regime anyway, so
# buckets stop at 16 and larger decode batches run eager.
BUCKETS = (1, 2, 4, 8, 16)


class DecodeGraphRunner:
    def __init__(
        self,
        model,
        kv_pool,
        device: str,
        scratch_block: int,
        max_blocks_per_seq: int,
        max_bucket: int = 256,
    ) -> None:
        self.model = model
        self.kv_pool = kv_pool
        self.device = device
        self.scratch_slot = scratch_block * kv_pool.block_size
        self.scratch_block = scratch_block
        self.max_blocks = max_blocks_per_seq
        self.buckets = [b for b in BUCKETS if b <= max_bucket]
        self._graphs: dict = {}
        self._pool = None  # shared graph memory pool

    def can_run(self, batch: ForwardBatch) -> bool:
        return (
            batch.num_prefill_seqs == 0
            and 0 < batch.num_decode_seqs <= self.buckets[-1]
            and batch.decode_block_tables is not None
            and batch.decode_block_tables.shape[1] <= self.max_blocks
        )

    def _bucket_for(self, n: int) -> int:
        for b in self.buckets:
            if n <= b:
                return b
        raise ValueError(n)

    # ------------------------------------------------------------------
    def _capture(self, B: int) -> dict:
        dev = self.device
        static = {
            "token_ids": torch.zeros(B, dtype=torch.long, device=dev),
            "positions": torch.zeros(B, dtype=torch.long, device=dev),
            "slot_mapping": torch.full(
                (B,), self.scratch_slot, dtype=torch.long, device=dev
            ),
            "block_tables": torch.full(
                (B, self.max_blocks), self.scratch_block, dtype=torch.int32, device=dev
            ),
            "kv_lens": torch.ones(B, dtype=torch.int32, device=dev),
            "sample_indices": torch.arange(B, dtype=torch.long, device=dev),
        }

        def make_batch() -> ForwardBatch:
            return ForwardBatch(
                token_ids=static["token_ids"],
                positions=static["positions"],
                slot_mapping=static["slot_mapping"],
                num_prefill_seqs=0,
                num_prefill_tokens=0,
                num_decode_seqs=B,
                decode_block_tables=static["block_tables"],
                decode_kv_lens=static["kv_lens"],
                sample_indices=static["sample_indices"],
            )

        # warm up twice on a side stream (allocator + rccl lazy init)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                with torch.inference_mode():
                    out = self.model.forward(make_batch(), self.kv_pool)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        graph = torch.cuda.CUDAGraph()
        kw = {"pool": self._pool} if self._pool is not None else {}
        with torch.inference_mode():
            with torch.cuda.graph(graph, **kw):
                logits = self.model.forward(make_batch(), self.kv_pool)
        if self._pool is None:
            self._pool = graph.pool()
        logger.info("captured decode graph for bucket B=%d", B)
        return {"graph": graph, "static": static, "logits": logits}

    # ------------------------------------------------------------------
    def entry_for(self, n: int) -> dict:
        """Get (or capture) the graph entry for the bucket covering n."""
        B = self._bucket_for(n)
        entry = self._graphs.get(B)
        if entry is None:
            entry = self._capture(B)
            entry["bucket"] = B
            self._graphs[B] = entry
        return entry

    def load_batch(self, entry: dict, batch: ForwardBatch) -> None:
        """Fill the entry's static inputs from a decode batch (padded,
        one H2D copy per input)."""
        n = batch.num_decode_seqs
        B = entry["static"]["token_ids"].shape[0]
        st = entry["static"]
        tid = torch.full((B,), 0, dtype=torch.long)
        tid[:n] = batch.token_ids
        pos = torch.zeros(B, dtype=torch.long)
        pos[:n] = batch.positions
        slots = torch.full((B,), self.scratch_slot, dtype=torch.long)
        slots[:n] = batch.slot_mapping
        tables = torch.full((B, self.max_blocks), self.scratch_block, dtype=torch.int32)
        w = batch.decode_block_tables.shape[1]
        tables[:n, :w] = batch.decode_block_tables
        kvl = torch.ones(B, dtype=torch.int32)
        kvl[:n] = batch.decode_kv_lens
        st["token_ids"].copy_(tid, non_blocking=True)
        st["positions"].copy_(pos, non_blocking=True)
        st["slot_mapping"].copy_(slots, non_blocking=True)
        st["block_tables"].copy_(tables, non_blocking=True)
        st["kv_lens"].copy_(kvl, non_blocking=True)

    def run(self, batch: ForwardBatch) -> torch.Tensor:
        n = batch.num_decode_seqs
        entry = self.entry_for(n)
        self.load_batch(entry, batch)
        entry["graph"].replay()
        return entry["logits"][:n]
