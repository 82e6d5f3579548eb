"""Expert-parallel all-to-all MoE test over gloo (world 2, CPU): the
EP-dispatched MoE layer must match the single-process dense routing."""

import multiprocessing as mp
import os

import pytest

pytestmark = pytest.mark.dist


def _worker(rank, world, port, tmpfile, out_q):
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
        from dts_amd.models.config import get_model_spec
        from dts_amd.models.mixtral import MoEMLP
        from dts_amd.parallel.tp import TPContext

        spec = get_model_spec("mixtral-tiny")  # 4 experts, top-2
        state = torch.load(tmpfile)
        tp = TPContext.from_world()
        moe = MoEMLP(spec, tp, dtype=torch.float32)
        with torch.no_grad():
            moe.router_w.copy_(state["router"])
            lo = rank * moe.experts_local
            moe.gate_up_w.copy_(state["gate_up"][lo : lo + moe.experts_local])
            moe.down_w.copy_(state["down"][lo : lo + moe.experts_local])
        out = moe(state["x"])
        if rank == 0:
            out_q.put(out)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_ep_matches_dense(tmp_path):
    import torch

    from dts_amd.models.config import get_model_spec
    from dts_amd.models.mixtral import MoEMLP
    from dts_amd.parallel.tp import TPContext

    torch.manual_seed(0)
    spec = get_model_spec("mixtral-tiny")
    dense = MoEMLP(spec, TPContext.single(), dtype=torch.float32)
    with torch.no_grad():
        dense.router_w.normal_(0, 0.2)
        dense.gate_up_w.normal_(0, 0.05)
        dense.down_w.normal_(0, 0.05)
    x = torch.randn(17, spec.hidden_size)
    ref = dense(x)

    f = tmp_path / "moe.pt"
    torch.save(
        {
            "router": dense.router_w.data,
            "gate_up": dense.gate_up_w.data,
            "down": dense.down_w.data,
            "x": x,
        },
        f,
    )
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, 2, 29600 + (os.getpid() * 4 + 3) % 800, str(f), q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    out = q.get(timeout=90)
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-5)
