"""The MI355X serving engine + the LocalBackend seam.

ServingEngine owns one model replica, its paged KV pool, the
continuous-batching scheduler and the sampler, and runs the step loop on a
dedicated thread (the asyncio search loop stays responsive for events —
SURVEY.md §7 hard-part 5: request queue + futures across the GIL).

LocalBackend implements the InferenceBackend protocol
(dts_amd/llm/backend.py) — the in-process replacement for the reference's
HTTPS client (ref client.py:153): chat messages are rendered to a
token-exact template, structured phases get a constrained-decoding
FormGuide (serving/structured.py), and the awaitable future resolves when
the scheduler finishes the sequence.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
import os
import re
import threading
from dataclasses import dataclass
from typing import Optional

import torch

from dts_amd.llm.errors import ContextLengthError
from dts_amd.llm.types import Completion, Message, SamplingParams, Usage
from dts_amd.models.config import ModelSpec, get_model_spec
from dts_amd.serving import structured
from dts_amd.serving.kv_cache import BlockManager, KVCachePool
from dts_amd.serving.sampler import Sampler
from dts_amd.serving.scheduler import Scheduler
from dts_amd.serving.sequence import Sequence
from dts_amd.serving.tokenizer import (
    ChatTemplate,
    HFChatTemplate,
    HFTokenizer,
    load_tokenizer,
)
from dts_amd.utils.logging import logger


@dataclass
class GenerationResult:
    token_ids: list
    text: str
    finish_reason: str
    prompt_tokens: int
    completion_tokens: int


class ServingEngine:
    def __init__(
        self,
        model_name: str = "llama-3-8b",
        device: Optional[str] = None,
        dtype: Optional[torch.dtype] = None,
        num_blocks: Optional[int] = None,
        kv_memory_bytes: Optional[int] = None,
        kv_frac: float = 0.8,
        block_size: int = 16,
        max_batch_tokens: int = 8192,
        max_running: int = 256,
        spec_k: Optional[int] = None,
        weight_seed: int = 0,
        model: Optional[object] = None,
        tokenizer_path: Optional[str] = None,
        weights_path: Optional[str] = None,
    ) -> None:
        self.spec: ModelSpec = get_model_spec(model_name)
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        if dtype is None:
            dtype = torch.bfloat16 if device != "cpu" else torch.float32
        self.device = device
        self.dtype = dtype

        if model is not None:
            self.model = model
        else:
            self.model = build_model(self.spec, dtype=dtype, device=device)
            if weights_path:
                if self.spec.arch == "llama":
                    from dts_amd.models.weights import load_llama_safetensors

                    load_llama_safetensors(self.model, weights_path)
                elif self.spec.arch == "mixtral":
                    from dts_amd.models.weights import load_mixtral_safetensors

                    load_mixtral_safetensors(self.model, weights_path)
                elif self.spec.arch == "gpt2":
                    from dts_amd.models.weights import load_gpt2_safetensors

                    load_gpt2_safetensors(self.model, weights_path)
                else:
                    raise NotImplementedError(
                        f"safetensors loading for arch {self.spec.arch!r} "
                        "— pass a pre-loaded model= instead"
                    )
            else:
                self.model.random_init(seed=weight_seed)

        kv_heads = getattr(self.model, "num_kv_heads_local", None)
        if kv_heads is None:
            kv_heads = self.spec.num_kv_heads
        if num_blocks is None:
            if kv_memory_bytes is None:
                if device.startswith("cuda"):
                    # sized from free memory AFTER this engine's weights
                    # loaded — callers wanting pressure control pass
                    # kv_frac instead of guessing absolute bytes
                    free, _total = torch.cuda.mem_get_info()
                    kv_memory_bytes = int(free * kv_frac)
                else:
                    kv_memory_bytes = 256 << 20  # CPU tests
            num_blocks = KVCachePool.blocks_for_memory(
                kv_memory_bytes,
                self.spec.num_layers,
                kv_heads,
                self.spec.head_dim,
                block_size,
                dtype_bytes=2 if dtype == torch.bfloat16 else 4,
            )
        self.kv_pool = KVCachePool(
            self.spec.num_layers,
            kv_heads,
            self.spec.head_dim,
            num_blocks,
            block_size,
            dtype=dtype,
            device=device,
        )
        # last block reserved as the hipGraph pad-row scratch target
        self.block_manager = BlockManager(max(1, num_blocks - 1), block_size)
        self._scratch_block = num_blocks - 1
        # prompt-lookup speculative decoding (serving/spec.py): k draft
        # rows per decode seq; exact-match verification keeps the output
        # stream identical, so it is on by default (DTS_SPEC_K=0 disables)
        if spec_k is None:
            spec_k = int(os.environ.get("DTS_SPEC_K", "4"))
        self.spec_k = spec_k
        use_native = os.environ.get("DTS_NATIVE_CORE", "1") != "0"
        self.scheduler = None
        if use_native:
            try:
                from dts_amd.serving.native_scheduler import NativeScheduler

                self.scheduler = NativeScheduler(
                    max(1, num_blocks - 1),
                    block_size,
                    max_batch_tokens,
                    max_running,
                    spec_k=spec_k,
                )
            except Exception as e:  # noqa: BLE001
                logger.warning("native core unavailable (%s); Python scheduler", e)
        if self.scheduler is None:
            self.scheduler = Scheduler(
                self.block_manager, max_batch_tokens, max_running, spec_k=spec_k
            )
        self.sampler = Sampler(device)
        # a real tokenizer JSON (HF `tokenizers` format) when given —
        # its vocab must fit the model's embedding width; else the
        # synthetic byte-level tokenizer at full vocab width
        self.tokenizer = load_tokenizer(self.spec.vocab_size, tokenizer_path)
        if isinstance(self.tokenizer, HFTokenizer):
            if self.tokenizer.vocab_size > self.spec.vocab_size:
                raise ValueError(
                    f"tokenizer vocab {self.tokenizer.vocab_size} exceeds "
                    f"model vocab {self.spec.vocab_size}"
                )
            self.template = HFChatTemplate(self.tokenizer)
        else:
            self.template = ChatTemplate(self.tokenizer)

        self._graph_runner = None
        self._chain = None
        if (
            device.startswith("cuda")
            and os.environ.get("DTS_NO_HIPGRAPH") != "1"
            and getattr(self.model, "graph_capturable", True)
        ):
            from dts_amd.serving.graph_runner import DecodeGraphRunner

            self._graph_runner = DecodeGraphRunner(
                self.model,
                self.kv_pool,
                device,
                scratch_block=self._scratch_block,
                max_blocks_per_seq=self.spec.max_position // block_size,
                max_bucket=min(256, max_running),
            )
            if os.environ.get("DTS_NO_CHAIN") != "1":
                from dts_amd.serving.chain import ChainRunner

                self._chain = ChainRunner(self._graph_runner, device)

        self._lock = threading.Lock()
        self._work = threading.Condition(self._lock)
        self._futures: dict = {}  # seq_id -> Future
        self._thread: Optional[threading.Thread] = None
        self._stop = False
        # stats
        self.steps = 0
        self.graph_steps = 0
        self.tokens_sampled = 0
        self.tokens_prefilled = 0
        self.t_schedule = 0.0
        self.t_forward_graph = 0.0
        self.t_forward_eager = 0.0
        self.t_sample = 0.0
        self.t_post = 0.0
        self.eager_decode_steps = 0
        self.prefill_steps = 0
        self.spec_draft_tokens = 0
        self.spec_accepted_tokens = 0
        self.chain_steps = 0
        self.chains = 0
        self.req_count = 0
        self.req_latency_sum = 0.0
        self.req_latency_max = 0.0

    # ------------------------------------------------------------------
    def submit_tokens(
        self,
        prompt_ids: list,
        params: SamplingParams,
        guide: Optional[object] = None,
        stream_cb: Optional[object] = None,
    ) -> concurrent.futures.Future:
        if not prompt_ids:
            # an empty prompt segfaulted the native core and raised a raw
            # IndexError in the Python scheduler — reject it cleanly
            raise ValueError("empty prompt: at least one token required")
        lo, hi = min(prompt_ids), max(prompt_ids)
        if lo < 0 or hi >= self.spec.vocab_size:
            # an out-of-vocab id crashes the embed gather mid-step (and
            # on GPU would leave a device-side assert behind)
            raise ValueError(
                f"prompt token id out of range [0, {self.spec.vocab_size}): "
                f"min={lo} max={hi}"
            )
        if guide is not None and hasattr(guide, "token_budget"):
            # a finite form defines its own output size; never let a
            # free-text phase budget truncate it mid-form
            need = guide.token_budget()
            if params.max_tokens < need:
                import dataclasses

                params = dataclasses.replace(params, max_tokens=need)
            # context guard must cover the WHOLE form: forced segments
            # extend the sequence in multi-token chunks, so a prompt that
            # fits but whose form does not would push positions past the
            # rope table mid-generation (observed as an index crash at
            # max_position on the llama-tiny CPU rehearsal)
            if len(prompt_ids) + need + 8 > self.spec.max_position:
                raise ContextLengthError(
                    f"prompt of {len(prompt_ids)} tokens + structured form "
                    f"of {need} exceeds max_position {self.spec.max_position}"
                )
        elif len(prompt_ids) + 16 > self.spec.max_position:
            raise ContextLengthError(
                f"prompt of {len(prompt_ids)} tokens exceeds max_position "
                f"{self.spec.max_position}"
            )
        seq = Sequence(tokens=list(prompt_ids), params=params)
        seq.guide = guide
        seq.stream_cb = stream_cb
        if guide is not None:
            forced = guide.initial_forced()
            seq.tokens.extend(forced)
            seq.output_tokens.extend(forced)
        fut: concurrent.futures.Future = concurrent.futures.Future()
        import time as _time

        seq.submit_ts = _time.perf_counter()  # type: ignore[attr-defined]
        with self._work:
            self._futures[seq.seq_id] = fut
            self.scheduler.add(seq)
            self._work.notify_all()
        return fut

    # ------------------------------------------------------------------
    def _fail_stuck(self) -> None:
        """Fail futures of never-fitting requests (call under _lock)."""
        if not self.scheduler.stuck:
            return
        from dts_amd.llm.errors import BackendError

        for seq in self.scheduler.stuck:
            fut = self._futures.pop(seq.seq_id, None)
            if fut is not None and not fut.done():
                fut.set_exception(
                    BackendError(
                        f"request of {len(seq.tokens)} tokens exceeds "
                        f"the KV pool ({self.block_manager.num_blocks}"
                        f"x{self.block_manager.block_size} tokens)"
                    )
                )
        self.scheduler.stuck.clear()

    @torch.inference_mode()
    def step(self) -> bool:
        """One scheduling + forward + sample step. Returns True if it ran."""
        with self._lock:
            batch = self.scheduler.schedule()
            self._fail_stuck()
        if batch is None:
            return False
        try:
            return self._execute(batch)
        except Exception as e:  # noqa: BLE001
            # a poisoned batch (bad token id, kernel error) must fail
            # loudly for ITS requests only — before this guard the
            # un-cleared in_flight flags live-locked the whole engine
            self._fail_batch(batch, e)
            return True

    def _fail_batch(self, batch, err: Exception) -> None:
        from dts_amd.llm.errors import BackendError

        logger.error("engine step failed; aborting its batch: %s", err)
        with self._lock:
            for seq in list(getattr(batch, "_scheduled", []) or []):
                fut = self._futures.pop(seq.seq_id, None)
                if fut is not None and not fut.done():
                    fut.set_exception(BackendError(f"engine step failed: {err}"))
                try:
                    self.scheduler.abort(seq)
                except Exception:  # noqa: BLE001 — best-effort cleanup
                    logger.warning("abort failed for seq %s", seq.seq_id)

    @torch.inference_mode()
    def _execute(self, batch, allow_chain: bool = True) -> bool:
        """Forward + sample + postprocess one scheduled batch (the TP
        driver calls this directly after broadcasting the batch)."""
        import time as _time

        t0 = _time.perf_counter()
        if allow_chain and self._chain is not None and self._chain.eligible(batch):
            if self._run_chain(batch):
                return True
        self.steps += 1
        use_graph = self._graph_runner is not None and self._graph_runner.can_run(
            batch
        )
        logits = None
        if use_graph:
            try:
                logits = self._graph_runner.run(batch)
            except RuntimeError as e:
                # a capture-unsafe op in the model must degrade to eager,
                # not fail every pending request (a mid-bench Mixtral
                # capture abort killed judge futures before this guard)
                if "captur" not in str(e).lower():
                    raise
                logger.warning(
                    "hipGraph capture failed (%s); disabling graphs for "
                    "this engine and running eager",
                    e,
                )
                self._graph_runner = None
                self._chain = None
                use_graph = False
        if use_graph:
            self.graph_steps += 1
            t1 = _time.perf_counter()
            self.t_forward_graph += t1 - t0
        else:
            dev_batch = batch.to(self.device) if self.device != "cpu" else batch
            logits = self.model.forward(dev_batch, self.kv_pool)
            t1 = _time.perf_counter()
            self.t_forward_eager += t1 - t0
            if batch.num_prefill_seqs:
                self.prefill_steps += 1
            else:
                self.eager_decode_steps += 1
        sampled_seqs = batch._sampled_seqs  # type: ignore[attr-defined]
        # NOTE on timer attribution: kernel launches above are async, so
        # t_forward_* measures launch/replay time only; the first device
        # sync happens inside sample(), so t_sample absorbs the GPU
        # execution wait of the whole step. Per-kernel truth lives in
        # rocprof (profiles/), not these host timers.
        tokens = (
            self.sampler.sample(
                logits, sampled_seqs, positions=getattr(batch, "_sample_pos", None)
            )
            if sampled_seqs
            else []
        )
        t2 = _time.perf_counter()
        self.t_sample += t2 - t1
        with self._lock:
            self.tokens_prefilled += batch.num_prefill_tokens
            row_groups = getattr(batch, "_row_groups", None)
            if row_groups is None:
                # one sampled row per seq (no speculation)
                self.tokens_sampled += len(tokens)
                for seq, tok in zip(sampled_seqs, tokens):
                    self._handle_sampled(seq, tok)
            else:
                spec_drafts = getattr(batch, "_spec_drafts", None) or {}
                r = 0
                for seq, n_rows in row_groups:
                    if n_rows == 1:
                        self.tokens_sampled += 1
                        self._handle_sampled(seq, tokens[r])
                    else:
                        draft = spec_drafts[seq.seq_id]
                        emitted = self._verify_and_emit(
                            seq, tokens[r : r + n_rows], draft
                        )
                        self.spec_draft_tokens += n_rows - 1
                        self.spec_accepted_tokens += emitted - 1
                        self.scheduler.set_accepted(seq, emitted)
                    r += n_rows
            # advance AFTER appends: speculative chunks must register
            # block content that exists, and finished seqs are skipped
            self.scheduler.advance_computed(batch)
        self.t_post += _time.perf_counter() - t2
        return True

    def _run_chain(self, batch) -> bool:
        """Chained device-resident decode (serving/chain.py): enqueue up
        to CHAIN_MAX graph replays with on-device token feedback, process
        tokens lagging the GPU, sync once. Returns False to fall back to
        the normal single-step path (reserve failure / too close to the
        context cap)."""
        import time as _time

        from dts_amd.serving.chain import CHAIN_MAX, DEPTH, MIN_CHAIN

        seqs = batch._sampled_seqs  # type: ignore[attr-defined]
        with self._lock:
            W = CHAIN_MAX
            for s in seqs:
                W = min(W, self.spec.max_position - 8 - len(s.tokens))
                W = min(W, s.params.max_tokens - s.num_generated + 1)
            if W < MIN_CHAIN:
                return False
            for s in seqs:
                if not self.scheduler.reserve_tokens(s, len(s.tokens) + W):
                    return False  # KV pressure: normal path handles it
            # reservation may have grown block tables — rebuild the rows
            # the statics will load (the batch's tensor predates reserve)
            tbls = [self.scheduler.block_table_of(s) for s in seqs]
            m = max(len(t) for t in tbls)
            batch.decode_block_tables = torch.tensor(
                [t + [0] * (m - len(t)) for t in tbls], dtype=torch.int32
            )
        t0 = _time.perf_counter()
        ctx = self._chain.prepare(batch, seqs)
        active = [True] * len(seqs)
        deferred: list = []  # (seq, reason) — finished AFTER the sync
        executed = 0
        processed = 0
        stop = False
        while not stop and executed < W:
            # keep the GPU at most DEPTH steps ahead, polling arrivals on
            # every launch so a new request waits O(DEPTH) steps, not O(W)
            while executed - processed < DEPTH and executed < W and not stop:
                self._chain.launch_step(ctx, executed)
                executed += 1
                if self.scheduler.waiting_count() > 0:
                    stop = True
            # catch up on whatever the GPU already finished (no blocking)
            progressed = False
            while processed < executed and self._chain.step_ready(processed):
                self._process_chain_step(ctx, seqs, active, deferred, processed)
                processed += 1
                progressed = True
            if not any(active) or self.scheduler.waiting_count() > 0:
                stop = True
            elif not progressed and executed - processed >= DEPTH:
                # pipeline full and nothing completed yet: block on the
                # oldest in-flight step instead of spinning
                self._chain.wait_step(processed)
        # drain: ONE real wait on the last outstanding step
        while processed < executed:
            self._chain.wait_step(processed)
            self._process_chain_step(ctx, seqs, active, deferred, processed)
            processed += 1
        with self._lock:
            # all in-flight KV writes are complete (last event synced) —
            # safe to release finished seqs' blocks now
            for seq, reason in deferred:
                self._finish(seq, reason)
            for s in seqs:
                self.scheduler.set_accepted(s, 0)
            self.scheduler.advance_computed(batch)
        self.steps += executed
        self.graph_steps += executed
        self.chain_steps += executed
        self.chains += 1
        self.t_forward_graph += _time.perf_counter() - t0
        return True

    def _process_chain_step(
        self, ctx: dict, seqs: list, active: list, deferred: list, i: int
    ) -> None:
        toks = self._chain.tokens_of(ctx, i)
        with self._lock:
            for j, seq in enumerate(seqs):
                if not active[j]:
                    continue  # finished earlier in the chain: garbage row
                tok = toks[j]
                self.tokens_sampled += 1
                self.scheduler.chain_advance(seq, tok)
                if seq.stream_cb is not None:
                    try:
                        seq.stream_cb([tok])
                    except Exception:  # noqa: BLE001
                        seq.stream_cb = None
                if tok in self.template.stop_token_ids:
                    active[j] = False
                    deferred.append((seq, "stop"))
                elif (
                    len(seq.tokens) + 8 >= self.spec.max_position
                    or seq.num_generated >= seq.params.max_tokens
                ):
                    active[j] = False
                    deferred.append((seq, "length"))

    def _verify_and_emit(self, seq: Sequence, toks: list, draft: list) -> int:
        """Exact-match speculative verification (serving/spec.py): emit
        sampled row j while sampled j-1 matched draft j-1 — every emitted
        token is the model's own sample conditioned on accepted history,
        so the stream equals non-speculative decoding token for token."""
        from dts_amd.serving.sequence import SeqStatus

        emitted = 0
        for j, tok in enumerate(toks):
            if j > 0 and toks[j - 1] != draft[j - 1]:
                break
            self.tokens_sampled += 1
            self._handle_sampled(seq, tok)
            emitted += 1
            if seq.status != SeqStatus.RUNNING:
                break  # finished (stop/length) mid-group: drop the rest
        return emitted

    def _handle_sampled(self, seq: Sequence, tok: int) -> None:
        params = seq.params
        self.scheduler.append_token(seq, tok)
        if seq.stream_cb is not None:
            try:
                seq.stream_cb([tok])
            except Exception:  # noqa: BLE001 — streaming must not kill decode
                seq.stream_cb = None
        if seq.guide is not None:
            forced = seq.guide.on_token(tok)
            if forced:
                self.scheduler.extend_tokens(seq, forced)
                if seq.stream_cb is not None:
                    try:
                        seq.stream_cb(list(forced))
                    except Exception:  # noqa: BLE001
                        seq.stream_cb = None
            if seq.guide.done():
                self._finish(seq, "stop")
            elif (
                len(seq.tokens) + 8 >= self.spec.max_position
                or seq.num_generated >= params.max_tokens
            ):
                self._finish(seq, "length")
            return
        if tok in self.template.stop_token_ids:
            self._finish(seq, "stop")
            return
        if (
            len(seq.tokens) + 8 >= self.spec.max_position
            or seq.num_generated >= params.max_tokens
        ):
            # hard context cap: positions beyond the rope table are invalid
            self._finish(seq, "length")

    def _finish(self, seq: Sequence, reason: str) -> None:
        import time as _time

        ts = getattr(seq, "submit_ts", None)
        if ts is not None:
            lat = _time.perf_counter() - ts
            self.req_count += 1
            self.req_latency_sum += lat
            self.req_latency_max = max(self.req_latency_max, lat)
        self.scheduler.finish(seq, reason)
        self.sampler.release(seq)
        fut = self._futures.pop(seq.seq_id, None)
        if fut is not None and not fut.done():
            out_ids = list(seq.output_tokens)
            # drop a trailing stop token from the text
            if out_ids and out_ids[-1] in self.template.stop_token_ids:
                out_ids = out_ids[:-1]
            fut.set_result(
                GenerationResult(
                    token_ids=out_ids,
                    text=self.tokenizer.decode(out_ids),
                    finish_reason=reason,
                    prompt_tokens=seq.num_prompt_tokens,
                    completion_tokens=len(seq.output_tokens),
                )
            )

    # ------------------------------------------------------------------
    def run_until_idle(self) -> None:
        """Synchronous drain (tests / bench warm-up)."""
        fruitless = 0
        while self.scheduler.has_work():
            if self.step():
                fruitless = 0
            else:
                # a no-batch step is not idle: it may have failed stuck
                # requests or preempted; re-schedule while work remains
                fruitless += 1
                if fruitless > 10000:
                    raise RuntimeError(
                        "engine live-lock: requests pending but the "
                        "scheduler produced no batch for 10000 steps"
                    )

    def _loop(self) -> None:
        while True:
            with self._work:
                while not self._stop and not self.scheduler.has_work():
                    self._work.wait(timeout=0.1)
                if self._stop:
                    return
            try:
                if not self.step():
                    # no batch this step but work remains (e.g. stuck
                    # drain): yield briefly instead of spinning hot
                    import time as _t

                    _t.sleep(0.001)
            except Exception as e:  # noqa: BLE001
                logger.exception("engine step failed: %s", e)
                with self._lock:
                    for seq in list(self.scheduler.running):
                        self.sampler.release(seq)
                        self.scheduler.abort(seq)
                    # fail EVERY pending future (waiting/held requests
                    # included) — a broken step means no one behind it
                    # will ever be served, and a silent hang is worse
                    # than a loud error
                    for sid, fut in list(self._futures.items()):
                        if not fut.done():
                            fut.set_exception(e)
                        self._futures.pop(sid, None)

    def start(self) -> None:
        if self._thread is None:
            self._stop = False
            self._thread = threading.Thread(target=self._loop, daemon=True)
            self._thread.start()

    def stop(self) -> None:
        with self._work:
            self._stop = True
            self._work.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    @property
    def cache_stats(self) -> dict:
        src = (
            self.scheduler
            if hasattr(self.scheduler, "cache_hit_tokens")
            else self.block_manager
        )
        return {
            "cache_hit_tokens": src.cache_hit_tokens,
            "cache_miss_tokens": src.cache_miss_tokens,
            "free_blocks": src.num_free(),
            "steps": self.steps,
            "graph_steps": self.graph_steps,
            "tokens_sampled": self.tokens_sampled,
            "tokens_prefilled": self.tokens_prefilled,
            "native_scheduler": type(self.scheduler).__name__ == "NativeScheduler",
            "preemptions": getattr(self.scheduler, "preemptions", 0),
            "requests": self.req_count,
            "req_latency_mean_s": round(
                self.req_latency_sum / max(1, self.req_count), 3
            ),
            "req_latency_max_s": round(self.req_latency_max, 3),
            "eager_decode_steps": self.eager_decode_steps,
            "prefill_steps": self.prefill_steps,
            "spec_draft_tokens": self.spec_draft_tokens,
            "spec_accepted_tokens": self.spec_accepted_tokens,
            "chain_steps": self.chain_steps,
            "chains": self.chains,
            "t_forward_graph_s": round(self.t_forward_graph, 2),
            "t_forward_eager_s": round(self.t_forward_eager, 2),
            "t_sample_s": round(self.t_sample, 2),
            "t_post_s": round(self.t_post, 2),
        }


def build_model(spec: ModelSpec, dtype, device, tp=None):
    if spec.arch == "llama":
        from dts_amd.models.llama import LlamaModel

        return LlamaModel(spec, tp=tp, dtype=dtype, device=device)
    if spec.arch == "gpt2":
        from dts_amd.models.gpt2 import GPT2Model

        return GPT2Model(spec, dtype=dtype, device=device)
    if spec.arch == "mixtral":
        from dts_amd.models.mixtral import MixtralModel

        return MixtralModel(spec, tp=tp, dtype=dtype, device=device)
    raise ValueError(f"unknown arch {spec.arch}")


# ---------------------------------------------------------------------------
# Backend seam
# ---------------------------------------------------------------------------

_N_RE = re.compile(r"exactly (\d+)")
_STRAT_SPLIT_RE = re.compile(r"This is strategy (\d+) of (\d+)")
_TRAJ_RE = re.compile(r"--- Trajectory ([0-9a-fA-F-]+)")


class LocalBackend:
    """InferenceBackend over one or more ServingEngines (per-phase models,
    ref engine.py:72-76 model resolution)."""

    def __init__(self, engines: dict, default_model: str) -> None:
        self.engines = engines
        self.default_model = default_model
        for e in engines.values():
            e.start()

    @classmethod
    def single(cls, engine: ServingEngine, name: str = "local") -> "LocalBackend":
        return cls({name: engine}, name)

    def shutdown(self) -> None:
        for e in self.engines.values():
            e.stop()

    def _engine(self, model: Optional[str]) -> ServingEngine:
        if model and model in self.engines:
            return self.engines[model]
        return self.engines[self.default_model]

    def _build_guide(self, engine: ServingEngine, messages: list):
        system = messages[0].content if messages and messages[0].role == "system" else ""
        system = system or ""
        user = next((m.content for m in reversed(messages) if m.role == "user"), "") or ""
        tok = engine.tokenizer
        if "[dts:strategy]" in system:
            ms = _STRAT_SPLIT_RE.search(user)
            if ms:  # split mode: one strategy per call, numbered keys
                return structured.strategy_form(tok, 1, start=int(ms.group(1)))
            m = _N_RE.search(user)
            return structured.strategy_form(tok, int(m.group(1)) if m else 4)
        if "[dts:intent]" in system:
            m = _N_RE.search(user)
            return structured.intent_form(tok, int(m.group(1)) if m else 3)
        if "[dts:judge-absolute]" in system:
            return structured.absolute_judge_form(tok)
        if "[dts:judge-comparative]" in system:
            tail = user[-160:]
            if "[dts:part=critique]" in tail:
                return structured.critique_form(tok)
            ids = _TRAJ_RE.findall(user)
            if "[dts:part=ranking]" in tail:
                return structured.ranking_only_form(tok, ids) if ids else None
            if ids:
                return structured.comparative_judge_form(tok, ids)
        return None

    async def chat(
        self,
        messages: list,
        params: SamplingParams,
        model: Optional[str] = None,
    ) -> Completion:
        engine = self._engine(model)
        prompt_ids = engine.template.render(messages)
        guide = self._build_guide(engine, messages) if params.json_mode else None
        fut = engine.submit_tokens(prompt_ids, params, guide=guide)
        result: GenerationResult = await asyncio.wrap_future(fut)
        if guide is not None and result.finish_reason == "length":
            # a structured form that ran out of context/budget produces
            # truncated JSON on EVERY retry — fail fast instead of letting
            # the parse-retry ladder regenerate it (ref client.py:148-203
            # would loop; locally the failure is deterministic)
            raise ContextLengthError(
                f"structured generation truncated at {result.completion_tokens} "
                f"tokens (prompt {result.prompt_tokens}); raise max_position "
                "or shrink the form"
            )
        return Completion(
            message=Message.assistant(result.text),
            usage=Usage(
                prompt_tokens=result.prompt_tokens,
                completion_tokens=result.completion_tokens,
            ),
            model=model or self.default_model,
            finish_reason=result.finish_reason,
        )

    async def stream(
        self,
        messages: list,
        params: SamplingParams,
        model: Optional[str] = None,
    ):
        """Async iterator of text deltas (parity: ref client.py:205-272
        `LLM.stream`). Tokens are pushed from the engine thread onto an
        asyncio queue; UTF-8 multibyte sequences are buffered until whole."""
        engine = self._engine(model)
        prompt_ids = engine.template.render(messages)
        loop = asyncio.get_running_loop()
        queue: asyncio.Queue = asyncio.Queue()

        def on_tokens(toks: list) -> None:
            loop.call_soon_threadsafe(queue.put_nowait, list(toks))

        fut = engine.submit_tokens(prompt_ids, params, stream_cb=on_tokens)
        afut = asyncio.wrap_future(fut)
        stops = set(engine.template.stop_token_ids)
        buf = bytearray()
        tok = engine.tokenizer

        if isinstance(tok, HFTokenizer):
            # incremental detokenization: decode the growing id list and
            # emit the new suffix — per-token decode would mangle
            # multi-token UTF-8 sequences, and ids < 256 are ordinary
            # vocab entries for an HF tokenizer, not raw bytes
            ids: list = []
            emitted = [0]  # chars already yielded

            def emit(toks: list) -> str:
                ids.extend(t for t in toks if t not in stops)
                text = tok.decode(ids)
                # hold back a trailing replacement char: it usually marks
                # an incomplete multibyte sequence that the next token
                # completes
                safe = len(text)
                while safe > emitted[0] and text[safe - 1] == "�":
                    safe -= 1
                delta = text[emitted[0] : safe]
                emitted[0] = safe
                return delta

        else:

            def emit(toks: list) -> str:
                # synthetic byte-level tokenizer: ids < 256 ARE raw UTF-8
                # bytes; buffer until a whole codepoint decodes
                parts = []
                for t in toks:
                    if t in stops:
                        continue
                    if t < 256:
                        buf.append(t)
                        try:
                            parts.append(buf.decode("utf-8"))
                            buf.clear()
                        except UnicodeDecodeError:
                            continue
                    else:
                        if buf:
                            parts.append(buf.decode("utf-8", errors="replace"))
                            buf.clear()
                        parts.append(tok.decode([t]))
                return "".join(parts)

        while True:
            get_task = asyncio.ensure_future(queue.get())
            done, _ = await asyncio.wait(
                {get_task, afut}, return_when=asyncio.FIRST_COMPLETED
            )
            if get_task in done:
                delta = emit(get_task.result())
                if delta:
                    yield delta
            else:
                get_task.cancel()
                while not queue.empty():
                    delta = emit(queue.get_nowait())
                    if delta:
                        yield delta
                if buf:  # trailing incomplete UTF-8 sequence
                    yield buf.decode("utf-8", errors="replace")
                    buf.clear()
                if isinstance(tok, HFTokenizer) and ids:
                    tail = tok.decode(ids)[emitted[0] :]
                    if tail:
                        yield tail
                await afut  # surface exceptions
                return
