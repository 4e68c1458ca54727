"""Expert parallelism: experts sharded across the TP group, token
exchange by all-to-all over xGMI (SURVEY §2.12 row '(absent) TP/PP/EP/SP';
BASELINE config 4: Mixtral tool-calling agent, expert all-to-all).

Layout: with ep_size = tp group size, rank r owns experts
[r*E/ep, (r+1)*E/ep) at FULL width (no intra-expert TP) — each token's
expert compute runs entirely on the owner rank, so the only collectives
are the two token exchanges. RCCL provides all_to_all_single over xGMI;
the gloo (CPU test) backend lacks all-to-all and uses point-to-point
sends — same math, test-only transport.
"""
from __future__ import annotations

import os
from typing import List

import torch
import torch.distributed as dist

from . import dist as pdist


def ep_enabled() -> bool:
    return (os.environ.get("AURORA_EP", "0") == "1"
            and pdist.tp_world_size() > 1)


def ep_size() -> int:
    return pdist.tp_world_size() if ep_enabled() else 1


def ep_rank() -> int:
    return pdist.tp_rank()


def all_to_all_tokens(x_send: torch.Tensor, send_splits: List[int],
                      recv_splits: List[int]) -> torch.Tensor:
    """Exchange row blocks: rank r receives recv_splits[s] rows from each
    rank s. x_send is ordered by destination rank."""
    group = pdist.tp_group()
    world = pdist.tp_world_size()
    out = x_send.new_empty(sum(recv_splits), x_send.shape[1])
    backend = dist.get_backend(group)
    if backend == "nccl":
        dist.all_to_all_single(out, x_send.contiguous(),
                               output_split_sizes=recv_splits,
                               input_split_sizes=send_splits, group=group)
        return out
    # gloo fallback: pairwise isend/irecv (correctness path for CPU tests)
    my = ep_rank()
    send_off = [0]
    for s in send_splits:
        send_off.append(send_off[-1] + s)
    recv_off = [0]
    for s in recv_splits:
        recv_off.append(recv_off[-1] + s)
    reqs = []
    # global rank of tp-rank i in my group
    base = dist.get_rank() - my
    for r in range(world):
        if r == my:
            out[recv_off[r]:recv_off[r + 1]] = \
                x_send[send_off[r]:send_off[r + 1]]
            continue
        if send_splits[r]:
            reqs.append(dist.isend(
                x_send[send_off[r]:send_off[r + 1]].contiguous(),
                dst=base + r, group=group))
        if recv_splits[r]:
            reqs.append(dist.irecv(out[recv_off[r]:recv_off[r + 1]],
                                   src=base + r, group=group))
    for q in reqs:
        q.wait()
    return out


def exchange_splits(send_splits: List[int]) -> List[int]:
    """recv_splits[s] = how many rows rank s sends to me."""
    group = pdist.tp_group()
    world = pdist.tp_world_size()
    t = torch.tensor(send_splits, dtype=torch.long)
    if torch.cuda.is_available() and dist.get_backend(group) == "nccl":
        t = t.cuda()
    gathered = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(gathered, t, group=group)
    me = ep_rank()
    return [int(g[me]) for g in gathered]


def ep_moe_forward(x: torch.Tensor, weights: torch.Tensor,
                   sel: torch.Tensor, n_experts: int,
                   expert_fn) -> torch.Tensor:
    """Dispatch -> remote expert compute -> combine.

    x: [T, H]; weights/sel: [T, k] routing; expert_fn(xp, local_sel)
    computes the owner-local experts on packed rows (local_sel holds
    LOCAL expert indices aligned with xp's rows) and returns [A_local, H].
    """
    T, H = x.shape
    k = sel.shape[1]
    world = ep_size()
    flat = sel.reshape(-1)  # [T*k] global expert ids
    # experts divide evenly over ranks (enforced at model build)
    e_per = n_experts // world
    owner = flat // e_per
    # sort assignments by (owner, expert) so sends are contiguous
    key = owner * (e_per * world) + flat
    order = key.argsort(stable=True)
    tok_of = order // k
    x_send = x[tok_of]  # [T*k, H]
    send_counts = torch.zeros(world, dtype=torch.long).scatter_add_(
        0, owner.cpu(), torch.ones_like(owner.cpu()))
    send_splits = [int(c) for c in send_counts]
    recv_splits = exchange_splits(send_splits)
    x_recv = all_to_all_tokens(x_send, send_splits, recv_splits)
    # local expert index per received row: exchange the expert ids too
    eid_send = (flat[order] % e_per).to(x.dtype).unsqueeze(1)
    eid_recv = all_to_all_tokens(eid_send, send_splits, recv_splits)
    local_sel = eid_recv.squeeze(1).to(torch.long)
    y_local = expert_fn(x_recv, local_sel)
    # return rows to their origin ranks (reverse exchange)
    y_back = all_to_all_tokens(y_local, recv_splits, send_splits)
    # combine on the origin: scale by routing weight, sum over k
    out = torch.zeros_like(x)
    wflat = weights.reshape(-1)[order].unsqueeze(1).to(x.dtype)
    out.index_add_(0, tok_of, y_back * wflat)
    return out
