"""Expert-parallel all-to-all token dispatch.

SURVEY.md §2.3 row "RCCL all-to-all (MoE expert dispatch)": with experts
sharded over the group, each (token, expert) assignment is shipped to the
rank owning that expert, computed there, and shipped back. On the xGMI
mesh this moves k/world of the activations per hop (vs a full all-reduce
of the combined output), and both hops are point-to-point — the traffic
pattern xGMI's 7 direct links serve best.

gloo (CPU tests) lacks all_to_all_single for this shape on some builds,
so a send/recv emulation backs the same interface.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def all_to_all_rows(
    x: torch.Tensor, send_splits: list, recv_splits: list, group=None
) -> torch.Tensor:
    """Exchange row-blocks of x: rank r receives recv_splits[r] rows from
    each peer according to their send_splits. Returns [sum(recv), H]."""
    out = torch.empty(
        (sum(recv_splits), *x.shape[1:]), dtype=x.dtype, device=x.device
    )
    backend = dist.get_backend(group)
    if backend == "nccl":
        dist.all_to_all_single(
            out, x.contiguous(), recv_splits, send_splits, group=group
        )
        return out
    # gloo emulation: pairwise isend/irecv (local copy for self)
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    send_offs = [0]
    for s in send_splits:
        send_offs.append(send_offs[-1] + s)
    recv_offs = [0]
    for s in recv_splits:
        recv_offs.append(recv_offs[-1] + s)
    x = x.contiguous()
    reqs = []
    for peer in range(world):
        if peer == rank:
            out[recv_offs[peer] : recv_offs[peer + 1]] = x[
                send_offs[peer] : send_offs[peer + 1]
            ]
            continue
        if send_splits[peer]:
            reqs.append(
                dist.isend(
                    x[send_offs[peer] : send_offs[peer + 1]], peer, group=group
                )
            )
        if recv_splits[peer]:
            reqs.append(
                dist.irecv(
                    out[recv_offs[peer] : recv_offs[peer + 1]], peer, group=group
                )
            )
    for r in reqs:
        r.wait()
    return out


def exchange_splits(send_splits: list, group=None) -> list:
    """All-to-all of the per-rank send counts -> per-rank recv counts."""
    world = dist.get_world_size(group)
    send = torch.tensor(send_splits, dtype=torch.long)
    recv = torch.empty(world, dtype=torch.long)
    backend = dist.get_backend(group)
    if backend == "nccl":
        dev = torch.device("cuda", torch.cuda.current_device())
        recv = recv.to(dev)
        dist.all_to_all_single(recv, send.to(dev), group=group)
        return [int(v) for v in recv.cpu()]
    gathered: list = [None] * world
    dist.all_gather_object(gathered, send_splits, group=group)
    rank = dist.get_rank(group)
    return [gathered[peer][rank] for peer in range(world)]
