"""Tensor-parallel serving: one model shard per rank, driver-scheduled.

BASELINE.json config 4 (Llama-3-70B TP=8 over xGMI): every rank holds a
head/expert shard (dts_amd/parallel/tp.py); rank 0 runs the full
ServingEngine (scheduler, KV bookkeeping, sampling) and broadcasts each
step's ForwardBatch to the worker ranks, which hold only their model shard
+ KV pool and replay the forward. The per-layer RCCL all-reduces inside
row-parallel linears synchronize the ranks implicitly; one extra object
broadcast per step carries the batch metadata.

Block tables/slots are identical on every rank (driver-owned), so each
rank's KV pool holds its own kv-head shard at the same block geometry.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from dts_amd.serving.batch import ForwardBatch
from dts_amd.serving.engine import ServingEngine
from dts_amd.utils.logging import logger

_STOP = "__tp_stop__"


def _batch_to_obj(batch: ForwardBatch) -> dict:
    def mv(t):
        return t.cpu() if t is not None else None

    return {
        "token_ids": mv(batch.token_ids),
        "positions": mv(batch.positions),
        "slot_mapping": mv(batch.slot_mapping),
        "num_prefill_seqs": batch.num_prefill_seqs,
        "num_prefill_tokens": batch.num_prefill_tokens,
        "cu_q": mv(batch.cu_q),
        "prefill_block_tables": mv(batch.prefill_block_tables),
        "prefill_kv_lens": mv(batch.prefill_kv_lens),
        "num_decode_seqs": batch.num_decode_seqs,
        "decode_block_tables": mv(batch.decode_block_tables),
        "decode_kv_lens": mv(batch.decode_kv_lens),
        "sample_indices": mv(batch.sample_indices),
    }


def _obj_to_batch(obj: dict) -> ForwardBatch:
    return ForwardBatch(**obj)


class TPDriverMixin:
    """Hooks installed on the rank-0 ServingEngine."""

    @staticmethod
    def install(engine: ServingEngine, group=None) -> None:
        orig_step = engine.step

        def step() -> bool:
            # schedule under lock, then broadcast + forward
            with engine._lock:
                batch = engine.scheduler.schedule()
                if engine.scheduler.stuck:
                    pass  # handled by orig path below via re-entry
            if batch is None:
                dist.broadcast_object_list([None], src=0, group=group)
                return False
            dist.broadcast_object_list([_batch_to_obj(batch)], src=0, group=group)
            return engine._tp_execute(batch)

        # reuse the engine's internals for forward+sample+postprocess
        def _tp_execute(batch) -> bool:
            import time as _time

            engine.steps += 1
            t0 = _time.perf_counter()
            dev_batch = batch.to(engine.device) if engine.device != "cpu" else batch
            with torch.inference_mode():
                logits = engine.model.forward(dev_batch, engine.kv_pool)
            t1 = _time.perf_counter()
            engine.t_forward_eager += t1 - t0
            sampled_seqs = batch._sampled_seqs
            tokens = engine.sampler.sample(logits, sampled_seqs) if sampled_seqs else []
            dist.broadcast_object_list([tokens], src=0, group=group)
            with engine._lock:
                engine.scheduler.advance_computed(batch)
                engine.tokens_prefilled += batch.num_prefill_tokens
                engine.tokens_sampled += len(tokens)
                for seq, tok in zip(sampled_seqs, tokens):
                    engine._handle_sampled(seq, tok)
            return True

        engine._tp_execute = _tp_execute
        engine.step = step
        engine._tp_orig_step = orig_step
        # graphs not yet wired through the broadcast path
        engine._graph_runner = None

    @staticmethod
    def shutdown(group=None) -> None:
        dist.broadcast_object_list([_STOP], src=0, group=group)


def run_tp_worker(model, kv_pool, device: str, group=None) -> None:
    """Worker loop for ranks > 0: replay driver batches until stop."""
    logger.info("TP worker rank %d ready", dist.get_rank(group))
    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0, group=group)
        obj = box[0]
        if obj is None:
            continue
        if obj == _STOP:
            return
        batch = _obj_to_batch(obj)
        dev_batch = batch.to(device) if device != "cpu" else batch
        with torch.inference_mode():
            model.forward(dev_batch, kv_pool)
        # consume the sampled-token broadcast (driver-side bookkeeping only)
        tok_box = [None]
        dist.broadcast_object_list(tok_box, src=0, group=group)
