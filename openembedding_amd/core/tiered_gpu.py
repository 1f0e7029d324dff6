"""GPU capacity tier v2: HBM row-cache over a pinned host-DRAM backing store.

BASELINE.json config 5 ("host-DRAM cold-row offload tier, async prefetch on
side stream") and the reference's PMem tier re-based on the MI355X memory
hierarchy (SURVEY §2.1 PmemEmbeddingTable / CacheItemPool):

  HBM (288 GB)  = the cache tier: HipVariableShard's hash table + row slab,
                  bounded at ``cache_rows``; per-slot last-touch work_id
                  stamps (the reference's LRU work_id, PmemEmbeddingTable.h)
  host DRAM     = the backing store: pinned slabs addressed by a SECOND
                  device-side hash table (key -> host slot), so admission,
                  fault-in and spill all run as HIP kernels reading/writing
                  the pinned memory directly (zero-copy: only touched rows
                  cross the host link) — no python dicts, no key lists.

Hot path: exactly HipVariableShard's sync-free bounded path plus one
fault-in kernel (no-op until something was spilled) and one touch-stamp
scatter. Eviction runs at the COMMIT boundary (end of update_weights),
after every pending gradient block was applied — so the table slots saved
in live pull handles are never invalidated mid-step (the v1 design evicted
at pull time and could corrupt prefetched handles). A slot-epoch guard
turns any residual stale-handle use into a loud error.

Driven through Variable.prefetch/pulling() the whole pull (fault-ins
included) runs on the prefetch side stream, which is what hides the host
link latency off the critical path.
"""

from __future__ import annotations

from typing import Optional

import torch

from .variable import VariableMeta
from .variable_gpu import HipVariableShard


class HipTieredVariableShard(HipVariableShard):
    HOST_INITIAL_ROWS = 1 << 14
    HOST_INITIAL_TABLE = 1 << 16

    def __init__(self, meta: VariableMeta, shard_id: int = 0,
                 shard_num: int = 1, device: str = "cuda", seed: int = 0,
                 cache_rows: int = 1 << 22):
        if not meta.use_hash_table:
            raise ValueError("the capacity tier requires hash mode")
        super().__init__(meta, shard_id, shard_num, device, seed)
        self.cache_rows = int(cache_rows)
        self.work_id = 0
        dev = self.device
        # +1: slot -1 maps to dummy index 0 so the stamp scatter never
        # branches on the host (sync-free)
        self._touch = torch.zeros(self.weights.shape[0] + 1,
                                  dtype=torch.int64, device=dev)
        # device-side host map: key -> host-slab slot
        self._host_cap = self.HOST_INITIAL_TABLE
        self.host_tk = torch.full((self._host_cap,), -1, dtype=torch.int64,
                                  device=dev)
        self.host_tv = torch.empty(self._host_cap, dtype=torch.int32,
                                   device=dev)
        self.host_nrows_dev = torch.zeros(1, dtype=torch.int32, device=dev)
        self.host_slot_keys = torch.empty(self.HOST_INITIAL_ROWS,
                                          dtype=torch.int64, device=dev)
        pin = torch.cuda.is_available()
        self._host_weights = torch.zeros((self.HOST_INITIAL_ROWS, self.dim),
                                         dtype=self.dtype, pin_memory=pin)
        self._host_state = torch.zeros((self.HOST_INITIAL_ROWS, 0),
                                       dtype=self.dtype, pin_memory=pin)
        self._host_rows_upper = 0     # conservative >= live host rows
        self._host_live = False       # False until the first spill
        self._faulted = torch.zeros(1, dtype=torch.int32, device=dev)
        self._evict_epoch = 0
        self._last_batch_upper = 0
        self._evict_streak = 0      # consecutive commits that evicted
        self._checkpoint_work_id = None
        self._cache_full_since_ckpt = False

    # ---------------------------------------------------------- host slabs

    def _host_ensure(self, rows: int) -> None:
        cap = self._host_weights.shape[0]
        if rows <= cap:
            return
        new_cap = max(rows, cap * 2)
        # fault-in/spill kernels hold raw pointers into the old slabs:
        # drain the device before swapping them out
        torch.cuda.synchronize()
        pin = torch.cuda.is_available()
        hw = torch.zeros((new_cap, self.dim), dtype=self.dtype,
                         pin_memory=pin)
        hw[:cap] = self._host_weights
        self._host_weights = hw
        hs = torch.zeros((new_cap, self._host_state.shape[1]),
                         dtype=self.dtype, pin_memory=pin)
        hs[:cap] = self._host_state
        self._host_state = hs
        sk = torch.empty(new_cap, dtype=torch.int64, device=self.device)
        sk[:self.host_slot_keys.shape[0]] = self.host_slot_keys
        self.host_slot_keys = sk

    def _host_maybe_rehash(self) -> None:
        if self._host_rows_upper * 2 <= self._host_cap:
            return
        new_cap = self._host_cap
        while self._host_rows_upper * 2 > new_cap:
            new_cap *= 2
        tk_new = torch.empty(new_cap, dtype=torch.int64, device=self.device)
        tv_new = torch.empty(new_cap, dtype=torch.int32, device=self.device)
        self.ext.ht_rehash(self.host_tk, self.host_tv, tk_new, tv_new)
        self.host_tk, self.host_tv = tk_new, tv_new
        self._host_cap = new_cap

    def set_optimizer(self, category: str, **cfg) -> None:
        super().set_optimizer(category, **cfg)
        if self._host_state.shape[1] != self.state_dim:
            if self._host_live:
                raise RuntimeError("cannot change optimizer state layout "
                                   "after rows were spilled to the host "
                                   "tier")
            pin = torch.cuda.is_available()
            self._host_state = torch.zeros(
                (self._host_weights.shape[0], self.state_dim),
                dtype=self.dtype, pin_memory=pin)

    def _ensure_rows(self, need: int) -> None:
        super()._ensure_rows(need)
        if self._touch.numel() < self.weights.shape[0] + 1:
            t = torch.zeros(self.weights.shape[0] + 1, dtype=torch.int64,
                            device=self.device)
            t[:self._touch.numel()] = self._touch
            self._touch = t

    def reserve_rows(self, rows: int) -> None:
        # CombinedEmbedding pre-sizes for its whole key space; the tier's
        # point is bounding HBM at cache_rows, so clamp the reservation
        super().reserve_rows(min(int(rows), self.cache_rows))

    # ------------------------------------------------------------ hot path

    def _lookup_or_insert(self, keys: torch.Tensor, u_dev=None):
        if self._in_graph_capture():
            raise RuntimeError("the capacity tier is not hipGraph-"
                               "capturable (evictions need host control); "
                               "run eager with prefetch overlap")
        slots, new_mask = super()._lookup_or_insert(keys, u_dev)
        if self._host_live:
            self.ext.fault_in(keys, u_dev, self.host_tk, self.host_tv,
                              self._host_weights,
                              self._host_state if self.state_dim else None,
                              self.weights, self.state, slots, new_mask,
                              self._faulted)
        # LRU stamp: slot -1 lands on dummy index 0 (sync-free)
        self._touch.scatter_(0, slots + 1, self.work_id)
        slots._oe_epoch = self._evict_epoch
        self._last_batch_upper = max(self._last_batch_upper, keys.numel())
        return slots, new_mask

    def push_slots(self, keys_buf, u_dev, slots, grads, counts) -> None:
        if getattr(slots, "_oe_epoch", self._evict_epoch) != self._evict_epoch:
            raise RuntimeError(
                "stale pull handle: the cache was compacted (eviction) "
                "after this pull; pull handles must be pushed before the "
                "commit that follows them")
        super().push_slots(keys_buf, u_dev, slots, grads, counts)

    def pull_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        slots = self._lookup_readonly(keys)
        empty = torch.empty(0, dtype=torch.uint8, device=self.device)
        out = self.ext.gather_init(self.weights, self.state, slots, empty,
                                   keys, None, 0, 0.0, 0.0, 0.0, self.seed,
                                   torch.empty(0, dtype=self.dtype,
                                               device=self.device), True,
                                   None)
        if self._host_live:
            self.ext.gather_host(keys, slots, self.host_tk, self.host_tv,
                                 self._host_weights, out)
        return out

    def update_weights(self) -> None:
        super().update_weights()
        self.work_id += 1
        # evict at the commit boundary: all pending gradient blocks were
        # just applied, so no live handle's slots are invalidated
        if self._nrows_upper + self._last_batch_upper > self.cache_rows:
            self._evict()
        else:
            self._evict_streak = 0

    # ------------------------------------------------------------- eviction

    def _sync_nrows(self) -> None:
        self._nrows_exact = int(self.nrows_dev.item())
        self._nrows_upper = self._nrows_exact

    def _spill(self, evict_slots: torch.Tensor) -> None:
        """Copy cache rows (by slot) into the host tier (insert keys into
        the device-side host map, then one spill kernel)."""
        n_ev = evict_slots.numel()
        if n_ev == 0:
            return
        keys = self.host_keys_of_slots(evict_slots)
        self._host_ensure(self._host_rows_upper + n_ev)
        # count the incoming rows BEFORE the load-factor check so the
        # insert below never runs into a >50%-full probe table
        self._host_rows_upper += n_ev
        self._host_maybe_rehash()
        hslots, _ = self.ext.ht_lookup(self.host_tk, self.host_tv, keys,
                                       self.host_nrows_dev,
                                       self.host_slot_keys, True, None)
        self.ext.spill_rows(evict_slots, hslots, self.weights, self.state,
                            self._host_weights,
                            self._host_state if self.state_dim else None)
        self._host_live = True
        # tighten the host-rows bound (one sync, eviction is off the hot
        # path by construction)
        self._host_rows_upper = int(self.host_nrows_dev.item())

    def host_keys_of_slots(self, slots: torch.Tensor) -> torch.Tensor:
        return self.slot_keys.index_select(0, slots)

    def _evict(self) -> None:
        self._sync_nrows()
        n = self._nrows_exact
        margin = self._last_batch_upper
        if n + margin <= self.cache_rows:
            return
        # steady thrash (working set >> cache: an eviction every commit):
        # evict deeper so the fixed rebuild cost amortizes over more
        # admitted rows (reference PmemEmbeddingTable evicts per-row and
        # has no rebuild; an epoch rebuild must batch harder instead)
        self._evict_streak += 1
        keep_frac = 4 if self._evict_streak >= 4 else (4 * 3)
        keep_target = max(1, (self.cache_rows * keep_frac) // 16 - margin)
        n_evict = n - keep_target
        if n_evict <= 0:
            return
        self._cache_full_since_ckpt = True
        touches = self._touch[1:n + 1]
        # exact count (not a threshold mask): with work_id ties — every row
        # touched every step — a mask would evict the whole cache
        evict_slots = torch.topk(touches, n_evict, largest=False,
                                 sorted=False).indices
        evict_mask = torch.zeros(n, dtype=torch.bool, device=self.device)
        evict_mask[evict_slots] = True
        self._spill(evict_slots)
        # rebuild the cache with the survivors, compacted
        keep_slots = (~evict_mask).nonzero(as_tuple=True)[0]
        kk = self.slot_keys.index_select(0, keep_slots)
        kw = self.weights.index_select(0, keep_slots)
        ks = (self.state.index_select(0, keep_slots)
              if self.state_dim else None)
        kt = touches.index_select(0, keep_slots)
        self.tk.fill_(-1)
        self.nrows_dev.zero_()
        self._nrows_upper = 0
        self._nrows_exact = 0
        m = keep_slots.numel()
        if m:
            slots, _ = self.ext.ht_lookup(self.tk, self.tv, kk,
                                          self.nrows_dev, self.slot_keys,
                                          True, None)
            self.weights[slots] = kw
            if ks is not None:
                self.state[slots] = ks
            self._touch.zero_()
            self._touch.scatter_(0, slots + 1, kt)
            self._nrows_exact = m
            self._nrows_upper = m
        else:
            self._touch.zero_()
        self._evict_epoch += 1

    # ----------------------------------------------------------- statistics

    def fault_count(self) -> int:
        """Rows faulted in from the host tier since start (one sync)."""
        return int(self._faulted.item())

    # ----------------------------------------------------------- persistence

    def should_persist(self) -> bool:
        return self._cache_full_since_ckpt and self._checkpoint_work_id is None

    def persist(self) -> int:
        """Lightweight checkpoint: flush every cached row to the host tier
        (reference persist; rows stay cached and keep training)."""
        self._sync_nrows()
        n = self._nrows_exact
        if n:
            self._spill(torch.arange(n, dtype=torch.int64,
                                     device=self.device))
        self._checkpoint_work_id = self.work_id
        self._cache_full_since_ckpt = False
        return self.work_id

    def checkpoint_committed(self) -> None:
        self._checkpoint_work_id = None

    def _host_only(self):
        """(keys, host_slots) of host-tier rows not currently cached."""
        hn = int(self.host_nrows_dev.item())
        if hn == 0:
            e = torch.empty(0, dtype=torch.int64, device=self.device)
            return e, e
        hkeys = self.host_slot_keys[:hn]
        cache_slots, _ = self.ext.ht_lookup(self.tk, self.tv, hkeys,
                                            self.nrows_dev, self.slot_keys,
                                            False, None)
        mask = cache_slots < 0
        return hkeys[mask], torch.arange(hn, device=self.device)[mask]

    @property
    def num_rows(self) -> int:
        n = int(self.nrows_dev.item())
        hkeys, _ = self._host_only()
        return n + hkeys.numel()

    def export_rows(self, include_state: bool = True):
        keys_c, w_c, s_c = super().export_rows(include_state)
        hkeys, hslots = self._host_only()
        if hkeys.numel() == 0:
            return keys_c, w_c, s_c
        hs_cpu = hslots.cpu()
        keys = torch.cat([keys_c, hkeys])
        w = torch.cat([w_c, self._host_weights[hs_cpu].to(self.device)])
        s = None
        if include_state and self.state_dim:
            s = torch.cat([s_c, self._host_state[hs_cpu].to(self.device)])
        return keys, w, s

    def clear(self) -> None:
        super().clear()
        self.host_tk.fill_(-1)
        self.host_nrows_dev.zero_()
        self._host_rows_upper = 0
        self._host_live = False
        self._touch.zero_()
        self._cache_full_since_ckpt = False
        self._checkpoint_work_id = None
        self._evict_epoch += 1
