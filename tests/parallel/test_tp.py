"""TP linear correctness over gloo (world 2, CPU): a column-parallel →
row-parallel pair must reproduce the unsharded computation."""

import multiprocessing as mp
import os

import pytest

pytestmark = pytest.mark.dist


def _worker(rank, world, port, out_q):
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
        from dts_amd.parallel.tp import (
            ColumnParallelLinear,
            RowParallelLinear,
            TPContext,
        )

        torch.manual_seed(0)
        H, I, T = 32, 64, 5
        # full reference weights (same on all ranks)
        w_up = torch.randn(I, H)
        w_down = torch.randn(H, I)
        x = torch.randn(T, H)

        tp = TPContext.from_world()
        col = ColumnParallelLinear(H, I, tp, dtype=torch.float32)
        row = RowParallelLinear(I, H, tp, dtype=torch.float32)
        shard = I // world
        with torch.no_grad():
            col.weight.copy_(w_up[rank * shard : (rank + 1) * shard])
            row.weight.copy_(w_down[:, rank * shard : (rank + 1) * shard])

        y = row(col(x))
        ref = x @ w_up.T @ w_down.T
        ok = torch.allclose(y, ref, atol=1e-4)

        # gather_output column-parallel
        col_g = ColumnParallelLinear(
            H, I, tp, dtype=torch.float32, gather_output=True
        )
        with torch.no_grad():
            col_g.weight.copy_(w_up[rank * shard : (rank + 1) * shard])
        yg = col_g(x)
        ok2 = torch.allclose(yg, x @ w_up.T, atol=1e-4)
        out_q.put((rank, bool(ok and ok2)))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_tp_pair_matches_dense():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29600 + (os.getpid() * 4 + 1) % 800, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=90) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    assert all(ok for _, ok in results)
