"""TP serving-engine test over gloo (world 2, CPU, fp32): the sharded
engine must reproduce the dense single-process engine's greedy decode."""

import multiprocessing as mp
import os

import pytest

pytestmark = pytest.mark.dist


def _worker(rank, world, port, tmpdir, out_q):
    os.environ.update(
        {
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        }
    )
    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from dts_amd.llm.types import SamplingParams
        from dts_amd.models.config import get_model_spec
        from dts_amd.models.llama import LlamaModel
        from dts_amd.models.weights import load_llama_safetensors
        from dts_amd.parallel.tp import TPContext
        from dts_amd.serving import ServingEngine
        from dts_amd.serving.engine import build_model
        from dts_amd.serving.kv_cache import KVCachePool
        from dts_amd.serving.tp_engine import TPDriverMixin, run_tp_worker

        spec = get_model_spec("llama-tiny")
        tp = TPContext.from_world()
        model = LlamaModel(spec, tp=tp, dtype=torch.float32, device="cpu")
        load_llama_safetensors(model, tmpdir)

        prompt = list(range(1, 40))
        if rank == 0:
            from dts_amd.serving.structured import strategy_form

            engine = ServingEngine(
                model_name="llama-tiny",
                device="cpu",
                dtype=torch.float32,
                num_blocks=256,
                block_size=8,
                model=model,
            )
            TPDriverMixin.install(engine)
            fut = engine.submit_tokens(
                prompt, SamplingParams(max_tokens=8, temperature=0.0, seed=0)
            )
            gfut = engine.submit_tokens(
                list(range(40, 80)),
                SamplingParams(max_tokens=4096, temperature=0.0, seed=0),
                guide=strategy_form(engine.tokenizer, 2),
            )
            engine.run_until_idle()
            res = fut.result(timeout=30)
            gtext = gfut.result(timeout=30).text
            TPDriverMixin.shutdown()
            out_q.put(("tp", res.token_ids, gtext))
        else:
            pool = KVCachePool(
                spec.num_layers,
                model.num_kv_heads_local,
                spec.head_dim,
                num_blocks=128,
                block_size=8,
                dtype=torch.float32,
                device="cpu",
            )
            run_tp_worker(model, pool, "cpu")
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_tp_engine_matches_dense(tmp_path):
    import torch

    from dts_amd.llm.types import SamplingParams
    from dts_amd.models.config import get_model_spec
    from dts_amd.models.llama import LlamaModel
    from dts_amd.models.weights import save_llama_safetensors
    from dts_amd.serving import ServingEngine

    # dense reference weights + output
    spec = get_model_spec("llama-tiny")
    dense = LlamaModel(spec, dtype=torch.float32, device="cpu")
    dense.random_init(seed=21)
    save_llama_safetensors(dense, str(tmp_path))
    from dts_amd.serving.structured import strategy_form

    engine = ServingEngine(
        model_name="llama-tiny",
        device="cpu",
        dtype=torch.float32,
        num_blocks=256,
        block_size=8,
        model=dense,
    )
    fut = engine.submit_tokens(
        list(range(1, 40)), SamplingParams(max_tokens=8, temperature=0.0, seed=0)
    )
    gfut = engine.submit_tokens(
        list(range(40, 80)),
        SamplingParams(max_tokens=4096, temperature=0.0, seed=0),
        guide=strategy_form(engine.tokenizer, 2),
    )
    engine.run_until_idle()
    dense_tokens = fut.result(timeout=30).token_ids
    dense_guided = gfut.result(timeout=30).text

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, 2, 29600 + (os.getpid() * 4 + 2) % 800, str(tmp_path), q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    tag, tp_tokens, tp_guided = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert tag == "tp"
    assert tp_tokens == dense_tokens
    # constrained JSON decoding over the TP broadcast path: same text
    import json

    assert json.loads(tp_guided) == json.loads(dense_guided)
