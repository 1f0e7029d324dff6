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
import os
from typing import Dict, List, Optional

import torch

from ..core.variable import VariableShard
from ..ops import dispatch as ops
from ..utils.metrics import REGISTRY, stage_timer
from . import comm


class _PaddedPlan:
    """Fixed-capacity wire plan of the sync-free padded all-to-all.

    One per (variable, element count). Every rank ships a [world, cap] key
    block padded with the reserved key -1 (defined miss everywhere:
    ops/csrc/embops.hip k_bucketize_pad header comment), so the multi-rank
    step needs ZERO device->host reads and is hipGraph-capturable. Buffers
    are preallocated so a captured graph replays against stable addresses.

    ``cap`` trades wire volume against skew headroom: per-peer unique
    counts concentrate around u/world for modulo routing, so
    cap = ceil(n * slack / world) (slack = OEAMD_A2A_SLACK, default 2)
    bounds the overflow probability; overflowed keys are counted in
    ``overflow`` and surface as a loud RuntimeError at the next host-side
    check (ShardedVariable.check_padded_overflow), never as silent zeros.
    """

    def __init__(self, world: int, n: int, device: torch.device):
        slack = float(os.environ.get("OEAMD_A2A_SLACK", "2.0"))
        cap = int(-(-n * slack // world))          # ceil
        cap = (cap + 255) // 256 * 256
        self.cap = max(1, min(n, cap))
        self.world = world
        total = world * self.cap
        self.send_keys = torch.full((total,), -1, dtype=torch.int64,
                                    device=device)
        self.send_src = torch.full((total,), -1, dtype=torch.int32,
                                   device=device)
        self.pos_of = torch.full((n,), -1, dtype=torch.int32, device=device)
        self.counts = torch.zeros(world, dtype=torch.int32, device=device)
        self.overflow = torch.zeros(1, dtype=torch.int32, device=device)
        self.overflow_host = 0      # CPU-fallback bucketize accumulates here


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
    # padded (fixed-capacity, zero-host-sync) multi-rank route:
    padded: bool = False
    plan: Optional["_PaddedPlan"] = None
    owner_cpu: Optional[tuple] = None   # torch-fallback owner state


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
        self._plans: Dict[int, Optional[_PaddedPlan]] = {}
        self._commits_since_check = 0

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

    def pull(self, indices: torch.Tensor, readonly: bool = False,
             field_offsets: torch.Tensor = None):
        """indices: int64 tensor of any shape -> (weights [*shape, dim], handle).

        ``field_offsets`` (optional, [n_fields] int64, indices then the RAW
        [batch, n_fields] field ids): fused into the GPU unique kernels —
        flat key = id + offset[field] — saving the broadcast-add launch and
        its intermediate every pull. Non-bounded paths fold it eagerly.

        Collective when world_size > 1: every rank must call it the same
        number of times per variable per step."""
        bounded_ok = (not readonly
                      and getattr(self.shard, "pull_bounded", None)
                      is not None)
        if field_offsets is not None and not bounded_ok:
            indices = indices + field_offsets
            field_offsets = None
        flat = indices.reshape(-1).to(torch.int64)
        n = flat.numel()
        self.stat_pull_indices += n
        remote = self.world_size > 1 or self._force_remote
        if not remote and bounded_ok:
            return self._pull_local_bounded(indices, flat, field_offsets)
        if remote and not readonly:
            if self._use_padded():
                r = self._pull_remote_padded(indices, flat, field_offsets)
                if r is not None:
                    return r
            if bounded_ok:
                return self._pull_remote_bounded(indices, flat,
                                                 field_offsets)
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

    def _pull_local_bounded(self, indices: torch.Tensor, flat: torch.Tensor,
                            field_offsets: torch.Tensor = None):
        """Single-GPU sync-free pull: bounded unique buffer + fused
        gather/init/scatter; zero host round-trips (hipGraph-capturable)."""
        ext = self.shard.ext
        uk_buf, inverse, u_dev = ext.unique_bounded(flat, field_offsets)
        out, slots = self.shard.pull_bounded(uk_buf, u_dev, inverse)
        h = PullHandle(shape=indices.shape, unique=uk_buf, inverse=inverse,
                       bounded=True, u_dev=u_dev, slots=slots)
        return out.view(*indices.shape, self.shard.dim), h

    def _pull_remote_bounded(self, indices: torch.Tensor,
                             flat: torch.Tensor,
                             field_offsets: torch.Tensor = None):
        """Multi-rank pull on the GPU engine with ONE host sync per step
        (the combined split-size read): dedup, owner bucketize and the
        sentinel-last ordering all run device-side off the bounded unique
        (the exact path paid a second sync for the tight unique count)."""
        world = self.world_size
        ext = self.shard.ext
        dev = flat.device
        uk_buf, inverse, u_dev = ext.unique_bounded(flat,
                                                    field_offsets)  # no sync
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

    # --------------------------------------------- padded (sync-free) route

    # padded-mode override for tests: None = auto (env OEAMD_PADDED_A2A,
    # else on iff the GPU bounded engine is present), True/False = forced
    _padded = None

    def _use_padded(self) -> bool:
        if self._padded is not None:
            return bool(self._padded)
        env = os.environ.get("OEAMD_PADDED_A2A", "auto")
        if env == "0":
            return False
        if env == "1":
            return True
        return getattr(self.shard, "pull_bounded", None) is not None

    def _get_padded_plan(self, n: int,
                         device: torch.device) -> Optional[_PaddedPlan]:
        # TWO plans per shape, rotated per pull: a pull's wire buffers
        # (send_src/pos_of) must survive until its push consumes them, and
        # one prefetched pull may be outstanding while the current batch
        # pushes — two live handles max under the issue-after-commit
        # prefetch discipline (deeper hand-rolled prefetch would need a
        # deeper rotation; pulling() never issues more than one ahead)
        pair = self._plans.get(n, False)
        if pair is None:
            return None
        if pair is False:
            import torch.distributed as dist
            if comm.dist_ready() and dist.get_world_size() > 1:
                # one-time agreement check: cap derives from n, so all
                # ranks must see the same element count for this variable
                # (the bench shape is rank-uniform; ragged shapes fall
                # back to the exact route). One host sync per (var, n).
                dev = device if comm.backend() == "nccl" else "cpu"
                t = torch.tensor([n, -n], dtype=torch.int64, device=dev)
                dist.all_reduce(t, op=dist.ReduceOp.MIN)
                if int(t[0]) != n or int(-t[1]) != n:
                    self._plans[n] = None
                    return None
            pair = [_PaddedPlan(self.world_size, n, device),
                    _PaddedPlan(self.world_size, n, device), 0]
            self._plans[n] = pair
        pair[2] ^= 1
        return pair[pair[2]]

    def _pull_remote_padded(self, indices: torch.Tensor,
                            flat: torch.Tensor,
                            field_offsets: torch.Tensor = None):
        """Multi-rank pull with ZERO host syncs: fixed [world, cap] key
        blocks padded with the reserved key -1 ride two equal-split
        all-to-alls; padding resolves to defined misses on the owner and is
        dropped on return. hipGraph-capturable end to end."""
        world = self.world_size
        dim = self.shard.dim
        dev = flat.device
        n = flat.numel()
        plan = self._get_padded_plan(n, dev)
        if plan is None:
            return None
        gpu = getattr(self.shard, "pull_bounded", None) is not None
        if gpu:
            ext = self.shard.ext
            uk_buf, inverse, u_dev = ext.unique_bounded(flat, field_offsets)
            ext.bucketize_pad(uk_buf, u_dev, world, plan.cap,
                              plan.send_keys, plan.send_src, plan.pos_of,
                              plan.counts, plan.overflow)
        else:
            uk_buf, inverse = ops.unique_inverse(flat)
            u_dev = None
            self._bucketize_pad_torch(uk_buf, plan)
        REGISTRY.add("pull_indices", n)
        REGISTRY.add("pull_wire_keys", plan.send_keys.numel())
        h = PullHandle(shape=indices.shape, unique=uk_buf, inverse=inverse,
                       u_dev=u_dev, padded=True, plan=plan)
        recv_keys = comm.all_to_all_equal(plan.send_keys, world)
        if gpu:
            uk2_buf, inv2, u2_dev = ext.unique_bounded(recv_keys)
            rows_back, slots2 = self.shard.pull_bounded(uk2_buf, u2_dev,
                                                        inv2)
            h.owner_unique = uk2_buf
            h.owner_inverse = inv2
            h.owner_u_dev = u2_dev
            h.owner_slots = slots2
        else:
            rows_back = self._owner_pull_padded_torch(recv_keys, h)
        rows_recv = comm.all_to_all_equal(rows_back, world)
        if gpu:
            out = ext.scatter_out(rows_recv, inverse, plan.pos_of, n)
        else:
            pos = plan.pos_of.index_select(0, inverse).long()
            out = rows_recv.index_select(0, pos.clamp(min=0))
            out = out.masked_fill((pos < 0).unsqueeze(1), 0.0)
        return out.view(*indices.shape, dim), h

    def _bucketize_pad_torch(self, uk: torch.Tensor,
                             plan: _PaddedPlan) -> None:
        """Pure-torch bucketize into the padded wire layout (CPU engine /
        gloo tests; same wire format as the HIP k_bucketize_pad)."""
        world, cap = plan.world, plan.cap
        plan.send_keys.fill_(-1)
        plan.send_src.fill_(-1)
        plan.pos_of.fill_(-1)
        nz = (uk >= 0).nonzero(as_tuple=True)[0]
        k = uk.index_select(0, nz)
        own = k % world
        order = torch.argsort(own, stable=True)
        k_s, nz_s, own_s = k[order], nz[order], own[order]
        cnt = torch.bincount(own_s, minlength=world)
        start = torch.cumsum(cnt, 0) - cnt
        pos = torch.arange(k_s.numel(), device=uk.device) - start[own_s]
        ok = pos < cap
        slot = (own_s * cap + pos)[ok]
        plan.send_keys[slot] = k_s[ok]
        plan.send_src[slot] = nz_s[ok].to(torch.int32)
        plan.pos_of[nz_s[ok]] = slot.to(torch.int32)
        plan.overflow_host += int((~ok).sum())

    def _owner_pull_padded_torch(self, recv_keys: torch.Tensor,
                                 h: PullHandle) -> torch.Tensor:
        """Owner side of the padded route on the CPU engine: mask the -1
        padding, exact-dedup the rest, gather (+lazy init), scatter back
        into wire positions (padding returns zeros)."""
        dim = self.shard.dim
        valid = recv_keys >= 0
        vk = recv_keys[valid]
        uk2, inv2 = ops.unique_inverse(vk)
        rows_back = torch.zeros(recv_keys.numel(), dim,
                                dtype=self.shard.dtype,
                                device=recv_keys.device)
        if uk2.numel():
            rows = self.shard.pull(uk2)
            rows_back[valid] = rows.index_select(0, inv2)
        h.owner_cpu = (valid, uk2, inv2)
        return rows_back

    def _push_padded(self, h: PullHandle, g: torch.Tensor) -> None:
        dim = self.shard.dim
        plan = h.plan
        world = self.world_size
        nbuf = h.unique.numel()
        ugrads, counts = ops.reduce_by_inverse(h.inverse, g, nbuf)
        if h.u_dev is not None:             # GPU engine
            ext = self.shard.ext
            send_p = ext.gather_pad(ugrads, counts, plan.send_src)
            recv_p = comm.all_to_all_equal(send_p, world)
            g2c, _ = ops.reduce_by_inverse(h.owner_inverse,
                                           recv_p.contiguous(),
                                           h.owner_unique.numel())
            g2, c2 = ext.split_payload(g2c, h.owner_u_dev)
            self.shard.push_slots(h.owner_unique, h.owner_u_dev,
                                  h.owner_slots, g2, c2)
            return
        payload = torch.cat([ugrads,
                             counts.to(ugrads.dtype).unsqueeze(1)], dim=1)
        src = plan.send_src.long()
        send_p = payload.index_select(0, src.clamp(min=0))
        send_p = send_p.masked_fill((src < 0).unsqueeze(1), 0.0)
        recv_p = comm.all_to_all_equal(send_p, world)
        valid, uk2, inv2 = h.owner_cpu
        if uk2.numel() == 0:
            return
        g2c = torch.zeros(uk2.numel(), dim + 1, dtype=recv_p.dtype,
                          device=recv_p.device)
        g2c.index_add_(0, inv2, recv_p[valid])
        g2 = g2c[:, :dim].contiguous()
        c2 = g2c[:, dim].round().to(torch.int64)
        self.shard.push(uk2, g2, c2)

    def check_padded_overflow(self) -> None:
        """Raise if any padded wire block ever overflowed its capacity.
        Called every 64 commits (outside graph capture) and explicitly by
        bench.py after the timed loop; overflow means some keys silently
        read zeros, so the run must fail loudly. Fix: raise OEAMD_A2A_SLACK
        (or disable with OEAMD_PADDED_A2A=0)."""
        for pair in self._plans.values():
            if pair is None:
                continue
            for plan in pair[:2]:
                ov = plan.overflow_host + int(plan.overflow.sum().item())
                if ov:
                    raise RuntimeError(
                        f"padded all-to-all overflow on variable "
                        f"{self.variable_id}: {ov} keys were dropped "
                        f"(bucket capacity {plan.cap}/peer). Raise "
                        f"OEAMD_A2A_SLACK (default 2.0) or set "
                        f"OEAMD_PADDED_A2A=0.")

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
        if h.padded:
            self._push_padded(h, g)
            return
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
        if self._plans:
            self._commits_since_check += 1
            if self._commits_since_check >= 64:
                capturing = (torch.cuda.is_available()
                             and torch.cuda.is_current_stream_capturing())
                if not capturing:
                    self._commits_since_check = 0
                    self.check_padded_overflow()

    # ------------------------------------------------------- whole-table pulls

    def pull_dense(self, start: int, count: int,
                   readonly: bool = False) -> torch.Tensor:
        """Materialize rows [start, start+count) of a bounded-vocab variable on
        every rank (export path; reference save_as_original_model bulk pull,
        exb.py:506-547 — which pulled through the TRAINING path, so missing
        rows instantiate their initializer values). Collective."""
        keys = torch.arange(start, start + count, dtype=torch.int64,
                            device=self.shard.device)
        out, _ = self.pull(keys, readonly=readonly)
        return out
