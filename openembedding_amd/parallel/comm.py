"""Collective shims over torch.distributed.

The sparse pull/push path needs variable-split all-to-all (the reference's
worker->server RPC fan-out, EmbeddingPullOperator.cpp:40-114, becomes ONE
all_to_all_v over xGMI). RCCL ("nccl" backend on ROCm) implements
all_to_all_single natively as p2p over the 7 xGMI links; the gloo backend
(CPU tests) lacks it, so a send/recv fallback is provided here.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.distributed as dist


def dist_ready() -> bool:
    return dist.is_available() and dist.is_initialized()


def backend() -> str:
    return dist.get_backend() if dist_ready() else ""


def all_to_all_lengths(send_counts: torch.Tensor) -> torch.Tensor:
    """Exchange per-peer element counts. send_counts int64 [world] (any device)
    -> recv_counts int64 [world] on the same device."""
    if not dist_ready():
        return send_counts.clone()  # world-1 (forced-remote tests)
    world = dist.get_world_size()
    if backend() == "gloo":
        cpu = send_counts.detach().cpu()
        gathered = [torch.zeros_like(cpu) for _ in range(world)]
        dist.all_gather(gathered, cpu)
        rank = dist.get_rank()
        out = torch.stack(gathered)[:, rank].contiguous()
        return out.to(send_counts.device)
    out = torch.empty_like(send_counts)
    dist.all_to_all_single(out, send_counts.contiguous())
    return out


def all_to_all_v(inp: torch.Tensor, in_splits: Sequence[int],
                 out_splits: Sequence[int]) -> torch.Tensor:
    """Variable all-to-all along dim 0. inp [sum(in_splits), ...] ->
    [sum(out_splits), ...]. Splits are python ints (host-known)."""
    if not dist_ready():
        return inp.clone()          # world-1 (forced-remote tests)
    world = dist.get_world_size()
    rank = dist.get_rank()
    trailing = list(inp.shape[1:])
    out = torch.empty([int(sum(out_splits))] + trailing, dtype=inp.dtype,
                      device=inp.device)
    if backend() != "gloo":
        dist.all_to_all_single(out, inp.contiguous(),
                               output_split_sizes=list(out_splits),
                               input_split_sizes=list(in_splits))
        return out
    if inp.is_cuda:
        # gloo transports CPU buffers only: stage D2H/H2D so the GPU engine
        # can run multi-process on one device (scripts/rccl_2rank_1gpu.py —
        # RCCL itself refuses two ranks on one GPU)
        res = all_to_all_v(inp.cpu(), in_splits, out_splits)
        return res.to(inp.device)
    # gloo fallback: batched isend/irecv (no alltoall in ProcessGroupGloo)
    in_offs = _offsets(in_splits)
    out_offs = _offsets(out_splits)
    reqs = []
    inp = inp.contiguous()
    # local copy
    if in_splits[rank]:
        out[out_offs[rank]:out_offs[rank] + out_splits[rank]] = \
            inp[in_offs[rank]:in_offs[rank] + in_splits[rank]]
    recv_bufs = {}
    for peer in range(world):
        if peer == rank:
            continue
        if out_splits[peer]:
            buf = torch.empty([out_splits[peer]] + trailing, dtype=inp.dtype,
                              device=inp.device)
            recv_bufs[peer] = buf
            reqs.append(dist.irecv(buf, src=peer))
        if in_splits[peer]:
            chunk = inp[in_offs[peer]:in_offs[peer] + in_splits[peer]].contiguous()
            reqs.append(dist.isend(chunk, dst=peer))
    for r in reqs:
        r.wait()
    for peer, buf in recv_bufs.items():
        out[out_offs[peer]:out_offs[peer] + out_splits[peer]] = buf
    return out


def all_to_all_equal(inp: torch.Tensor, world: int) -> torch.Tensor:
    """Fixed-shape all-to-all: inp [world*cap, ...], chunk i of ``cap`` rows
    goes to rank i; returns the same shape. No split sizes on the wire -> no
    host sync, and the collective is hipGraph-capturable on nccl/RCCL."""
    if not dist_ready():
        return inp.clone()          # world-1 (forced-remote tests)
    if backend() != "gloo":
        out = torch.empty_like(inp)
        dist.all_to_all_single(out, inp.contiguous())
        return out
    cap = inp.shape[0] // world
    splits = [cap] * world
    return all_to_all_v(inp, splits, splits)


def _offsets(splits: Sequence[int]) -> List[int]:
    offs = [0]
    for s in splits:
        offs.append(offs[-1] + int(s))
    return offs


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    """Shim kept for API symmetry with the gloo-aware helpers above
    (dense-grad reduction itself lives in torch/__init__.py where it
    overlaps the sparse commit)."""
    if dist_ready():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t
