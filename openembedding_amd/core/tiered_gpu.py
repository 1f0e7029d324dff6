"""GPU capacity tier: HBM row-cache over a pinned host-DRAM backing store.

BASELINE.json config 5 ("host-DRAM cold-row offload tier, async hipMemcpy
prefetch on side stream") and the reference's PMem tier re-based on the
MI355X memory hierarchy (SURVEY §2.1 PmemEmbeddingTable / CacheItemPool):

  HBM (288 GB)  = the cache tier: HipVariableShard's hash table + row slab,
                  bounded at ``cache_rows``; per-slot last-touch work_id
                  stamps (the reference's LRU work_id, PmemEmbeddingTable.h)
  host DRAM     = the backing store: pinned tensors, rows move H2D/D2H with
                  non_blocking copies (hipMemcpyAsync under the hood — on
                  the prefetch side stream when driven through
                  Variable.prefetch/pulling, which is what hides the
                  fault-in latency off the critical path)

Key-set bookkeeping (which keys are spilled) is host-side — the fault-in
decision needs host knowledge anyway because the batch's key list arrives
from the host. The device never pays for it: the hot path (all keys cached)
is exactly HipVariableShard's sync-free path plus one touch-stamp write.

Eviction rebuilds the device hash table from the surviving keys (an
open-addressed table has no cheap delete); evictions happen at batch
granularity and move >= cache_rows/4 rows, so the rebuild amortizes.
"""

from __future__ import annotations

from typing import Dict

import torch

from .variable import VariableMeta
from .variable_gpu import HipVariableShard


class HipTieredVariableShard(HipVariableShard):
    def __init__(self, meta: VariableMeta, shard_id: int = 0, shard_num: int = 1,
                 device: str = "cuda", seed: int = 0,
                 cache_rows: int = 1 << 22):
        if not meta.use_hash_table:
            raise ValueError("the capacity tier requires hash mode")
        super().__init__(meta, shard_id, shard_num, device, seed)
        self.cache_rows = int(cache_rows)
        self.work_id = 0
        self._touch = torch.zeros(self.weights.shape[0], dtype=torch.int64,
                                  device=self.device)
        self._host_weights = torch.zeros((0, self.dim), dtype=self.dtype)
        self._host_state = torch.zeros((0, 0), dtype=self.dtype)
        self._host_free: list = []
        self._host_next = 0
        self._host_of: Dict[int, int] = {}
        self._checkpoint_work_id = None
        self._cache_full_since_ckpt = False
        self._last_slots = None

    # ------------------------------------------------------------ host slab

    def _host_grow(self, need: int) -> None:
        cap = self._host_weights.shape[0]
        if need <= cap:
            return
        new_cap = max(need, max(4096, cap * 2))
        pin = self.device.type == "cuda" and torch.cuda.is_available()
        hw = torch.zeros((new_cap, self.dim), dtype=self.dtype, pin_memory=pin)
        hw[:cap] = self._host_weights
        self._host_weights = hw
        hs = torch.zeros((new_cap, self._host_state.shape[1]),
                         dtype=self.dtype, pin_memory=pin)
        hs[:cap] = self._host_state
        self._host_state = hs

    def _host_alloc(self, key: int) -> int:
        hs = self._host_of.get(key)
        if hs is None:
            hs = self._host_free.pop() if self._host_free else self._host_next
            if hs == self._host_next:
                self._host_next += 1
            self._host_grow(self._host_next)
            self._host_of[key] = hs
        return hs

    def set_optimizer(self, category: str, **cfg) -> None:
        super().set_optimizer(category, **cfg)
        if self._host_state.shape[1] != self.state_dim:
            n = self._host_weights.shape[0]
            pin = self.device.type == "cuda" and torch.cuda.is_available()
            hs = torch.zeros((n, self.state_dim), dtype=self.dtype,
                             pin_memory=pin)
            if self._host_of and self.state_dim:
                used = torch.tensor(sorted(self._host_of.values()),
                                    dtype=torch.int64)
                row = self._make_state_init_row().cpu()
                hs[used] = row.expand(used.numel(), -1)
            self._host_state = hs

    def _ensure_rows(self, need: int) -> None:
        super()._ensure_rows(need)
        if self._touch.numel() < self.weights.shape[0]:
            t = torch.zeros(self.weights.shape[0], dtype=torch.int64,
                            device=self.device)
            t[:self._touch.numel()] = self._touch
            self._touch = t

    # -------------------------------------------------- fault-in / eviction

    def _tier_admit(self, keys: torch.Tensor) -> None:
        """Host-side batch admission: fault spilled keys back into the HBM
        cache and evict cold rows when over budget. Runs BEFORE the normal
        HIP pull; one host sync (keys.tolist) — hidden by the prefetch
        stream when driven through pulling()."""
        kl = keys.tolist()
        from_host = [k for k in dict.fromkeys(kl) if k in self._host_of]
        self._sync_nrows()
        need = self._nrows_exact + len(kl)  # upper bound of new cache rows
        if need > self.cache_rows:
            self._evict(set(kl))
        if from_host:
            hslots = torch.tensor([self._host_of[k] for k in from_host],
                                  dtype=torch.int64)
            kt = torch.tensor(from_host, dtype=torch.int64,
                              device=self.device)
            w = self._host_weights[hslots].to(self.device, non_blocking=True)
            s = (self._host_state[hslots].to(self.device, non_blocking=True)
                 if self.state_dim else None)
            slots, _ = self._lookup_or_insert(kt)
            self.weights[slots] = w
            if s is not None:
                self.state[slots] = s
            for k in from_host:
                self._host_free.append(self._host_of.pop(k))

    def _sync_nrows(self) -> None:
        self._nrows_exact = int(self.nrows_dev.item())
        self._nrows_upper = self._nrows_exact

    def _evict(self, pinned_keys) -> None:
        """Move the coldest rows (not in ``pinned_keys``) to host and rebuild
        the device table from the survivors."""
        self._cache_full_since_ckpt = True
        n = self._nrows_exact
        if n == 0:
            return
        n_evict_target = max(n - (3 * self.cache_rows) // 4,
                             self.cache_rows // 4)
        order = torch.argsort(self._touch[:n], stable=True)
        keys_sorted = self.slot_keys[:n].index_select(0, order).cpu()
        evict, evict_pos = [], []
        for pos, k in enumerate(keys_sorted.tolist()):
            if len(evict) >= n_evict_target:
                break
            if k in pinned_keys:
                continue
            evict.append(k)
            evict_pos.append(pos)
        if not evict:
            return
        evict_slots = order[torch.tensor(evict_pos, dtype=torch.int64,
                                         device=self.device)]
        w = self.weights[evict_slots].cpu()
        s = self.state[evict_slots].cpu() if self.state_dim else None
        for i, k in enumerate(evict):
            hs = self._host_alloc(k)
            self._host_weights[hs] = w[i]
            if s is not None:
                self._host_state[hs] = s[i]
        # rebuild the cache with the survivors, compacted
        mask = torch.ones(n, dtype=torch.bool, device=self.device)
        mask[evict_slots] = False
        keep = mask.nonzero(as_tuple=True)[0]
        kw = self.weights[keep].clone()
        ks = self.state[keep].clone() if self.state_dim else None
        kk = self.slot_keys[:n][keep].clone()
        kt = self._touch[:n][keep].clone()
        m = keep.numel()
        self.tk.fill_(-1)
        self.nrows_dev.zero_()
        self._nrows_upper = 0
        self._nrows_exact = 0
        if m:
            slots, _ = self.ext.ht_lookup(self.tk, self.tv, kk,
                                          self.nrows_dev, self.slot_keys,
                                          True, None)
            self.weights[slots] = kw
            if ks is not None:
                self.state[slots] = ks
            self._touch[slots] = kt
            self._nrows_exact = m
            self._nrows_upper = m

    # -------------------------------------------------------------- training

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        self._tier_admit(keys)
        out = super().pull(keys)
        slots = self._last_slots
        if slots is not None:
            # stamp only resolved slots: a miss slot (-1) must not corrupt
            # row 0's LRU order
            valid = slots >= 0
            self._touch[slots[valid]] = self.work_id
        return out

    def _lookup_or_insert(self, keys: torch.Tensor, u_dev=None):
        slots, new_mask = super()._lookup_or_insert(keys, u_dev)
        self._last_slots = slots
        return slots, new_mask

    # bounded-path opt-out: the tier needs the host key list per batch, so
    # the sharded engine must route through the exact path (it checks
    # `getattr(shard, "pull_bounded", None)`)
    pull_bounded = None
    push_slots = None

    def pull_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        out = super().pull_readonly(keys)
        kl = keys.tolist()
        rows = [(i, self._host_of[k]) for i, k in enumerate(kl)
                if k in self._host_of]
        if rows:
            idx = torch.tensor([i for i, _ in rows], dtype=torch.int64,
                               device=self.device)
            hs = torch.tensor([h for _, h in rows], dtype=torch.int64)
            out[idx] = self._host_weights[hs].to(self.device)
        return out

    def update_weights(self) -> None:
        super().update_weights()
        self.work_id += 1

    # ----------------------------------------------------------- persistence

    def should_persist(self) -> bool:
        return self._cache_full_since_ckpt and self._checkpoint_work_id is None

    def persist(self) -> int:
        self._sync_nrows()
        n = self._nrows_exact
        if n:
            keys = self.slot_keys[:n].cpu().tolist()
            w = self.weights[:n].cpu()
            s = self.state[:n].cpu() if self.state_dim else None
            for sl, k in enumerate(keys):
                hs = self._host_alloc(k)
                self._host_weights[hs] = w[sl]
                if s is not None:
                    self._host_state[hs] = s[sl]
        self._checkpoint_work_id = self.work_id
        self._cache_full_since_ckpt = False
        return self.work_id

    def checkpoint_committed(self) -> None:
        self._checkpoint_work_id = None

    @property
    def num_rows(self) -> int:
        cached = set(self.slot_keys[:int(self.nrows_dev.item())]
                     .cpu().tolist())
        return len(cached) + sum(1 for k in self._host_of if k not in cached)

    def export_rows(self, include_state: bool = True):
        keys_c, w_c, s_c = super().export_rows(include_state)
        cached = set(keys_c.cpu().tolist())
        items = sorted(((k, h) for k, h in self._host_of.items()
                        if k not in cached), key=lambda kv: kv[1])
        if not items:
            return keys_c, w_c, s_c
        hkeys = torch.tensor([k for k, _ in items], dtype=torch.int64,
                             device=self.device)
        hslots = torch.tensor([h for _, h in items], dtype=torch.int64)
        keys = torch.cat([keys_c, hkeys])
        w = torch.cat([w_c, self._host_weights[hslots].to(self.device)])
        s = None
        if include_state and self.state_dim:
            s = torch.cat([s_c, self._host_state[hslots].to(self.device)])
        return keys, w, s

    def clear(self) -> None:
        super().clear()
        self._host_of = {}
        self._host_free = []
        self._host_next = 0
        self._cache_full_since_ckpt = False
        self._checkpoint_work_id = None
