"""Mixtral through the TP/EP serving engine over gloo (world 2, CPU):
the sharded engine (attention TP + experts EP via all-to-all) must
reproduce the dense single-process engine's greedy decode — the
BASELINE config-5 multi-GPU shape."""

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
        from dts_amd.models.mixtral import MixtralModel
        from dts_amd.models.weights import load_mixtral_safetensors
        from dts_amd.parallel.tp import TPContext
        from dts_amd.serving import ServingEngine
        from dts_amd.serving.kv_cache import KVCachePool
        from dts_amd.serving.tp_engine import TPDriverMixin, run_tp_worker

        spec = get_model_spec("mixtral-tiny")
        tp = TPContext.from_world()
        model = MixtralModel(spec, tp=tp, dtype=torch.float32, device="cpu")
        load_mixtral_safetensors(model, tmpdir)

        if rank == 0:
            engine = ServingEngine(
                model_name="mixtral-tiny",
                device="cpu",
                dtype=torch.float32,
                num_blocks=128,
                block_size=8,
                model=model,
            )
            TPDriverMixin.install(engine)
            fut = engine.submit_tokens(
                list(range(1, 40)),
                SamplingParams(max_tokens=8, temperature=0.0, seed=0),
            )
            engine.run_until_idle()
            res = fut.result(timeout=30)
            TPDriverMixin.shutdown()
            out_q.put(("ep", res.token_ids))
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
def test_mixtral_ep_engine_matches_dense(tmp_path):
    import torch

    from dts_amd.llm.types import SamplingParams
    from dts_amd.models.config import get_model_spec
    from dts_amd.models.mixtral import MixtralModel
    from dts_amd.models.weights import save_mixtral_safetensors
    from dts_amd.serving import ServingEngine

    spec = get_model_spec("mixtral-tiny")
    dense = MixtralModel(spec, dtype=torch.float32, device="cpu")
    dense.random_init(seed=31)
    save_mixtral_safetensors(dense, str(tmp_path))
    engine = ServingEngine(
        model_name="mixtral-tiny",
        device="cpu",
        dtype=torch.float32,
        num_blocks=128,
        block_size=8,
        model=dense,
    )
    fut = engine.submit_tokens(
        list(range(1, 40)), SamplingParams(max_tokens=8, temperature=0.0, seed=0)
    )
    engine.run_until_idle()
    dense_tokens = fut.result(timeout=30).token_ids

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(
            target=_worker,
            args=(r, 2, 29600 + (os.getpid() * 4 + 3) % 800, str(tmp_path), q),
        )
        for r in range(2)
    ]
    for p in procs:
        p.start()
    tag, ep_tokens = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert tag == "ep"
    assert ep_tokens == dense_tokens
