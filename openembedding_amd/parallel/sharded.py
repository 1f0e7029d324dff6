"""ShardedVariable: the distributed pull/push engine.

This file collapses the reference's whole RPC data path —
EmbeddingPullOperator / EmbeddingPushOperator / EmbeddingStoreOperator +
client handles (reference openembedding/server/*.cpp, client/
EmbeddingVariableHandle.cpp) — into RCCL all-to-all over xGMI:

  pull  = unique+bucketize -> all_to_all_v(keys) -> owner gather (+lazy init)
          -> all_to_all_v(rows) -> scatter to duplicated positions
  push  = reduce-by-key locally (the reference's client-side pre-aggregation,
          EmbeddingPushOperator.cpp:39-58 — dedup before the wire is the key
          bandwidth win) -> all_to_all_v(grads,counts) -> owner merge
  commit= local fused optimizer apply, no communication
          (reference EmbeddingStoreOperator.cpp:23-81; ordering that the
          reference built from batch-id pending queues comes free from
          stream ordering of collectives).

Owner of global key k is rank ``k % world_size`` (reference shard routing,
EmbeddingPullOperator.cpp:74-78, kept for checkpoint compatibility).
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional

import torch

from ..core.variable import VariableShard
from ..ops import dispatch as ops
from ..utils.metrics import REGISTRY, stage_timer
from . import comm


@dataclasses.dataclass
class PullHandle:
    """Routing state captured by pull, reused by the matching push
    (the reference kept it as saved block offsets in the pull handler,
    EmbeddingPullOperator.cpp:67-79,229-249)."""

    shape: torch.Size
    unique: torch.Tensor            # [u] unique keys, local order
    inverse: torch.Tensor           # [n] position -> unique id
    order: Optional[torch.Tensor] = None       # [u] perm grouping unique by owner
    send_splits: Optional[List[int]] = None    # keys sent to each rank
    recv_splits: Optional[List[int]] = None    # keys received from each rank
    owner_unique: Optional[torch.Tensor] = None  # [u2] deduped keys owned here
    owner_inverse: Optional[torch.Tensor] = None  # recv pos -> owner_unique id
    # bounded (sync-free) single-GPU fast path:
    bounded: bool = False
    u_dev: Optional[torch.Tensor] = None       # device int32 live-count
    slots: Optional[torch.Tensor] = None       # table slots saved from pull
    # bounded owner-side state of the remote path (GPU multi-rank): the
    # owner's dedup/gather runs sync-free and its table slots are reused by
    # the matching push
    owner_u_dev: Optional[torch.Tensor] = None
    owner_slots: Optional[torch.Tensor] = None


class ShardedVariable:
    """User-facing variable handle; wraps the local shard + collectives."""

    def __init__(self, shard: VariableShard, storage=None):
        self.shard = shard
        self.storage = storage
        self.world_size = shard.shard_num
        self.rank = shard.shard_id
        # perf counters (reference pull_indices/pull_unique accumulators,
        # EmbeddingPullOperator.cpp:208-247 — they diagnose all-to-all sizing)
        self.stat_pull_indices = 0
        self.stat_pull_unique = 0

    # -------------------------------------------------------------- properties

    @property
    def meta(self):
        return self.shard.meta

    @property
    def variable_id(self):
        return self.shard.meta.variable_id

    @property
    def embedding_dim(self):
        return self.shard.dim

    def set_initializer(self, category: str, **cfg):
        self.shard.set_initializer(category, **cfg)

    def set_optimizer(self, category: str, **cfg):
        self.shard.set_optimizer(category, **cfg)

    def reserve_rows(self, total_rows: int) -> None:
        """Pre-size this shard for its share of ``total_rows`` keys (hash
        mode; makes the GPU insert path hipGraph-capturable)."""
        fn = getattr(self.shard, "reserve_rows", None)
        if fn is not None:
            fn((int(total_rows) + self.world_size - 1) // self.world_size)

    # ------------------------------------------------------------------- pull

    # test hook: run the multi-rank path at world_size 1 (collectives become
    # identity) so the GPU bucketize/fan-out logic is validatable on one GPU
    _force_remote = False

    def pull(self, indices: torch.Tensor, readonly: bool = False):
        """indices: int64 tensor of any shape -> (weights [*shape, dim], handle).

        Collective when world_size > 1: every rank must call it the same
        number of times per variable per step."""
        flat = indices.reshape(-1).to(torch.int64)
        n = flat.numel()
        self.stat_pull_indices += n
        remote = self.world_size > 1 or self._force_remote
        if (not remote and not readonly
                and getattr(self.shard, "pull_bounded", None) is not None):
            return self._pull_local_bounded(indices, flat)
        if (remote and not readonly
                and getattr(self.shard, "pull_bounded", None) is not None):
            return self._pull_remote_bounded(indices, flat)
        unique, inverse = ops.unique_inverse(flat)
        self.stat_pull_unique += unique.numel()
        # reference pull_indices/pull_unique accumulators
        # (EmbeddingPullOperator.cpp:208-247): dedup rate sizes the wire
        REGISTRY.add("pull_indices", n)
        REGISTRY.add("pull_unique", unique.numel())
        h = PullHandle(shape=indices.shape, unique=unique, inverse=inverse)
        if self.world_size == 1:
            rows_u = (self.shard.pull_readonly(unique) if readonly
                      else self.shard.pull(unique))
        else:
            rows_u = self._pull_remote(h, readonly)
        out = rows_u.index_select(0, inverse)
        out = out.view(*h.shape, self.shard.dim)
        return out, h

    def _pull_local_bounded(self, indices: torch.Tensor, flat: torch.Tensor):
        """Single-GPU sync-free pull: bounded unique buffer + fused
        gather/init/scatter; zero host round-trips (hipGraph-capturable)."""
        ext = self.shard.ext
        uk_buf, inverse, u_dev = ext.unique_bounded(flat)
        out, slots = self.shard.pull_bounded(uk_buf, u_dev, inverse)
        h = PullHandle(shape=indices.shape, unique=uk_buf, inverse=inverse,
                       bounded=True, u_dev=u_dev, slots=slots)
        return out.view(*indices.shape, self.shard.dim), h

    def _pull_remote_bounded(self, indices: torch.Tensor,
                             flat: torch.Tensor):
        """Multi-rank pull on the GPU engine with ONE host sync per step
        (the combined split-size read): dedup, owner bucketize and the
        sentinel-last ordering all run device-side off the bounded unique
        (the exact path paid a second sync for the tight unique count)."""
        world = self.world_size
        ext = self.shard.ext
        dev = flat.device
        uk_buf, inverse, u_dev = ext.unique_bounded(flat)   # no sync
        n = uk_buf.numel()
        valid = torch.arange(n, device=dev) < u_dev.to(torch.int64)
        owner = torch.where(valid, uk_buf % world,
                            torch.full_like(uk_buf, world))
        send_counts = torch.bincount(owner, minlength=world + 1)[:world]
        recv_counts = comm.all_to_all_lengths(send_counts)
        both = torch.cat([send_counts, recv_counts]).tolist()  # ONE sync
        send_splits, recv_splits = both[:world], both[world:]
        u = int(sum(send_splits))
        self.stat_pull_unique += u
        REGISTRY.add("pull_indices", n)
        REGISTRY.add("pull_unique", u)
        order = torch.argsort(owner, stable=True)[:u]  # valid, owner-grouped
        send_keys = uk_buf.index_select(0, order)
        recv_keys = comm.all_to_all_v(send_keys, send_splits, recv_splits)
        h = PullHandle(shape=indices.shape, unique=uk_buf, inverse=inverse,
                       order=order, send_splits=send_splits,
                       recv_splits=recv_splits, u_dev=u_dev)
        if recv_keys.numel():
            uk2_buf, inv2, u2_dev = ext.unique_bounded(recv_keys)
            rows_back, slots2 = self.shard.pull_bounded(uk2_buf, u2_dev,
                                                        inv2)
            h.owner_unique = uk2_buf
            h.owner_inverse = inv2
            h.owner_u_dev = u2_dev
            h.owner_slots = slots2
        else:
            h.owner_unique = recv_keys
            rows_back = torch.empty(0, self.shard.dim, dtype=torch.float32,
                                    device=dev)
        rows_sorted = comm.all_to_all_v(rows_back, recv_splits, send_splits)
        rows_u = torch.zeros(n, self.shard.dim, dtype=rows_sorted.dtype,
                             device=dev)
        rows_u.index_copy_(0, order, rows_sorted)
        out = rows_u.index_select(0, inverse)
        return out.view(*indices.shape, self.shard.dim), h

    def _pull_remote(self, h: PullHandle, readonly: bool) -> torch.Tensor:
        with stage_timer("pull", "remote"):
            return self._pull_remote_impl(h, readonly)

    def _pull_remote_impl(self, h: PullHandle, readonly: bool) -> torch.Tensor:
        world = self.world_size
        owner = h.unique % world
        order = torch.argsort(owner, stable=True)
        send_keys = h.unique.index_select(0, order)
        send_counts = torch.bincount(owner, minlength=world)
        recv_counts = comm.all_to_all_lengths(send_counts)
        send_splits = send_counts.tolist()
        recv_splits = recv_counts.tolist()
        recv_keys = comm.all_to_all_v(send_keys, send_splits, recv_splits)
        # owner side: dedup across ranks, gather (+lazy init), fan back out.
        # On the GPU engine this runs the bounded sync-free path (fused
        # dedup+gather+duplicate-scatter, slots kept for the push).
        if (not readonly
                and getattr(self.shard, "pull_bounded", None) is not None
                and recv_keys.numel()):
            uk2_buf, inv2, u2_dev = self.shard.ext.unique_bounded(recv_keys)
            rows_back, slots2 = self.shard.pull_bounded(uk2_buf, u2_dev, inv2)
            h.owner_unique = uk2_buf
            h.owner_inverse = inv2
            h.owner_u_dev = u2_dev
            h.owner_slots = slots2
        else:
            uk2, inv2 = ops.unique_inverse(recv_keys)
            rows_u2 = (self.shard.pull_readonly(uk2) if readonly
                       else self.shard.pull(uk2))
            rows_back = rows_u2.index_select(0, inv2)
            h.owner_unique = uk2
            h.owner_inverse = inv2
        rows_sorted = comm.all_to_all_v(rows_back, recv_splits, send_splits)
        rows_u = torch.empty_like(rows_sorted)
        rows_u.index_copy_(0, order, rows_sorted)
        h.order = order
        h.send_splits = send_splits
        h.recv_splits = recv_splits
        return rows_u

    # ------------------------------------------------------------------- push

    def push(self, h: PullHandle, grads: torch.Tensor) -> None:
        """grads: [*shape, dim] gradient of the pulled weights."""
        dim = self.shard.dim
        g = grads.reshape(-1, dim)
        u = h.unique.numel()
        ugrads, counts = ops.reduce_by_inverse(h.inverse, g, u)
        if h.bounded:
            self.shard.push_slots(h.unique, h.u_dev, h.slots, ugrads, counts)
            return
        if h.send_splits is None:       # pure-local (world 1) exact path
            self.shard.push(h.unique, ugrads, counts)
            return
        # one fused payload [u, dim+1] = grads ‖ counts-as-f32: a single
        # all_to_all instead of two (counts <= batch size, exact in fp32),
        # and the owner reduces grads+counts in ONE reduce-by-key pass
        payload = torch.cat([ugrads, counts.to(ugrads.dtype).unsqueeze(1)],
                            dim=1)
        send_p = payload.index_select(0, h.order)
        recv_p = comm.all_to_all_v(send_p, h.send_splits, h.recv_splits)
        u2 = h.owner_unique.numel()
        if u2 == 0:
            return  # no keys owned here this step
        g2c, _ = ops.reduce_by_inverse(h.owner_inverse,
                                       recv_p.contiguous(), u2)
        g2 = g2c[:, :dim].contiguous()
        c2 = g2c[:, dim].round().to(torch.int64)
        if h.owner_slots is not None:
            # bounded owner path: slots saved from pull, sync-free apply
            self.shard.push_slots(h.owner_unique, h.owner_u_dev,
                                  h.owner_slots, g2, c2)
        else:
            self.shard.push(h.owner_unique, g2, c2)

    # ----------------------------------------------------------------- commit

    def update_weights(self) -> None:
        self.shard.update_weights()

    # ------------------------------------------------------- whole-table pulls

    def pull_dense(self, start: int, count: int) -> torch.Tensor:
        """Materialize rows [start, start+count) of a bounded-vocab variable on
        every rank (export path; reference save_as_original_model bulk pull,
        exb.py:506-547). Collective."""
        keys = torch.arange(start, start + count, dtype=torch.int64,
                            device=self.shard.device)
        out, _ = self.pull(keys, readonly=True)
        return out
