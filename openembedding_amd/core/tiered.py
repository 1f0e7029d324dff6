"""Tiered variable shard: device row-cache over a host-DRAM backing store.

MI355X rebuild of the reference's PMem tier (SURVEY §2.1:
PmemEmbeddingTable.h / PmemEmbeddingItemPool.h / PersistManager.h and the
paper's cache+pipeline design):

  reference                      | here
  -------------------------------+----------------------------------------
  DRAM cache over PMem rows      | HBM cache over host-DRAM rows
  ItemPointer tag cache-vs-pmem  | _loc per key: CACHE slot or host slot
  LRU list + work_id stamps      | per-slot last-touch batch id (work_id)
  flush on evict (pmem_flush)    | D2H copy of evicted rows (hipMemcpyAsync
                                 |   via non_blocking copy on pinned memory)
  cache budget (PersistManager)  | cache_rows budget (server.cache_size_mb)
  should_persist when cache full | same signal, same semantics
  checkpoint = work_id watermark | persist() stamps a checkpoint work_id;
    + data already persistent    |   rows at/below it live on host; dump
                                 |   writes config + watermark only

Semantics: identical to the plain hash shard — pull creates+initializes
missing rows, update applies the optimizer once per touched key.  The tier
only moves WHERE a row lives; tests assert bit-equality with an untired
shard under the same gradient stream (the reference's pmem tests do the
same, variable/pmem_embedding_table_test.cpp).

The eviction batch is 1/4 of the cache (coldest first) so the expensive
host round-trip amortizes — the reference evicts per-item but prefetches in
batches of 64 (PmemEmbeddingItemPool.h:32-129); on a GPU the batch form is
the only one that makes sense.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from .variable import VariableMeta, VariableShard


class TieredVariableShard(VariableShard):
    """Hash-mode shard whose device slab is bounded; cold rows spill to a
    host-memory backing store. Device-agnostic torch implementation (the
    CPU "device" tier is still exercised by tests; on cuda the cache slabs
    live in HBM and the backing store in pinned host memory)."""

    def __init__(self, meta: VariableMeta, shard_id: int = 0, shard_num: int = 1,
                 device: str = "cpu", seed: int = 0,
                 cache_rows: int = 1 << 20):
        if not meta.use_hash_table:
            raise ValueError("TieredVariableShard requires hash mode "
                             "(bounded-vocab tables fit HBM by construction)")
        super().__init__(meta, shard_id, shard_num, device, seed)
        self.cache_rows = int(cache_rows)
        self._pin = self.device.type == "cuda"
        # host backing store (grow-on-demand slab)
        self._host_weights = torch.zeros((0, self.dim), dtype=self.dtype)
        self._host_state = torch.zeros((0, 0), dtype=self.dtype)
        self._host_free: list = []
        self._host_next = 0                  # allocated host slots
        self._host_of: Dict[int, int] = {}   # key -> host slot (spilled rows)
        # per-cache-slot metadata
        self._touch = torch.zeros(0, dtype=torch.int64)  # last-touch work_id
        self.work_id = 0
        # checkpoint watermark (reference pending-checkpoint machinery,
        # PmemEmbeddingOptimizerVariable.h:47-86)
        self._checkpoint_work_id: Optional[int] = None
        self._cache_full_since_ckpt = False

    # ------------------------------------------------------------ host slab

    def _host_grow(self, need: int) -> None:
        cap = self._host_weights.shape[0]
        if need <= cap:
            return
        new_cap = max(need, max(1024, cap * 2))
        hw = torch.zeros((new_cap, self.dim), dtype=self.dtype,
                         pin_memory=self._pin)
        hw[:cap] = self._host_weights
        self._host_weights = hw
        hs = torch.zeros((new_cap, self._host_state.shape[1]),
                         dtype=self.dtype, pin_memory=self._pin)
        hs[:cap] = self._host_state
        self._host_state = hs

    def set_optimizer(self, category: str, **cfg) -> None:
        super().set_optimizer(category, **cfg)
        if self._host_state.shape[1] != self.state_dim:
            n = self._host_weights.shape[0]
            hs = torch.zeros((n, self.state_dim), dtype=self.dtype,
                             pin_memory=self._pin)
            if self._host_of and self.state_dim:
                used = torch.tensor(sorted(self._host_of.values()),
                                    dtype=torch.int64)
                row = self._make_state_init_row().cpu()
                hs[used] = row.expand(used.numel(), -1)
            self._host_state = hs

    def _ensure_rows(self, need: int) -> None:
        super()._ensure_rows(need)
        if self._touch.numel() < self.weights.shape[0]:
            t = torch.zeros(self.weights.shape[0], dtype=torch.int64)
            t[:self._touch.numel()] = self._touch
            self._touch = t

    def _host_alloc(self, key: int) -> int:
        """Host slot for ``key`` (reuse its existing copy's slot if any)."""
        hs = self._host_of.get(key)
        if hs is None:
            hs = self._host_free.pop() if self._host_free else self._host_next
            if hs == self._host_next:
                self._host_next += 1
            self._host_grow(self._host_next)
            self._host_of[key] = hs
        return hs

    # ------------------------------------------------------------- eviction

    def _evict_if_needed(self, incoming: int, pinned=frozenset()) -> None:
        """Make room for ``incoming`` new cache rows; evict coldest quarter
        when the budget would overflow (flush weights+state to host).
        ``pinned`` keys (the batch being pulled) are never evicted."""
        if self._nrows + incoming <= self.cache_rows:
            return
        self._cache_full_since_ckpt = True
        n_evict = max(self._nrows + incoming - self.cache_rows,
                      self.cache_rows // 4)
        # slot -> key reverse map (host-side dict scan; the GPU backend keeps
        # slot_keys for this)
        key_of = {s: k for k, s in self._index.items()}
        order = torch.argsort(self._touch[:self._nrows], stable=True)
        cand = [int(s) for s in order.tolist() if key_of[int(s)] not in pinned]
        n_evict = min(n_evict, len(cand))
        if n_evict == 0:
            return
        evict_slots = torch.tensor(cand[:n_evict], dtype=torch.int64)
        evict_keys = [key_of[int(s)] for s in evict_slots.tolist()]
        # flush rows to host
        w = self.weights[evict_slots].to("cpu")
        s = self.state[evict_slots].to("cpu") if self.state_dim else None
        for i, k in enumerate(evict_keys):
            hs = self._host_alloc(k)
            self._host_weights[hs] = w[i]
            if s is not None:
                self._host_state[hs] = s[i]
            del self._index[k]
        # compact the cache: move surviving rows down into evicted slots
        evict_set = set(evict_slots.tolist())
        keep = torch.tensor([s for s in range(self._nrows)
                             if s not in evict_set], dtype=torch.int64)
        if keep.numel():
            perm = torch.cat([keep, evict_slots])
            self.weights[:self._nrows] = self.weights[:self._nrows][perm]
            if self.state_dim:
                self.state[:self._nrows] = self.state[:self._nrows][perm]
            self._touch[:self._nrows] = self._touch[:self._nrows][perm]
            new_slot = {int(old): new for new, old in enumerate(perm.tolist())}
            self._index = {k: new_slot[s] for k, s in self._index.items()}
        self._nrows -= n_evict

    # -------------------------------------------------------------- lookups

    def _lookup_or_insert(self, keys: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Cache lookup with host-tier faulting: keys spilled to host are
        copied back into cache (the reference's pull done-callback cache
        population, PmemEmbeddingOptimizerVariable.h:179-192)."""
        kl = keys.tolist()
        seen = set()
        missing = [k for k in kl if k not in self._index
                   and not (k in seen or seen.add(k))]
        from_host = [k for k in missing if k in self._host_of]
        self._evict_if_needed(len(missing), pinned=frozenset(kl))
        if from_host:
            hslots = torch.tensor([self._host_of[k] for k in from_host],
                                  dtype=torch.int64)
            w = self._host_weights[hslots].to(self.device, non_blocking=True)
            s = (self._host_state[hslots].to(self.device, non_blocking=True)
                 if self.state_dim else None)
            need = self._nrows + len(from_host)
            self._ensure_rows(need)
            for i, k in enumerate(from_host):
                slot = self._nrows
                self._index[k] = slot
                self._nrows += 1
                self._host_free.append(self._host_of.pop(k))
            base = self._nrows - len(from_host)
            self.weights[base:self._nrows] = w
            if s is not None:
                self.state[base:self._nrows] = s
        # base class creates+initializes the truly-new rows
        slots, new_mask = super()._lookup_or_insert(keys)
        # (a batch alone bigger than the cache may overshoot the budget;
        # allowed, like the reference's reserved cache — trimmed at the next
        # batch boundary by the next _evict_if_needed)
        self._touch[slots.cpu()] = self.work_id
        return slots, new_mask

    def pull_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        out = super().pull_readonly(keys)
        kl = keys.tolist()
        rows = [(i, self._host_of[k]) for i, k in enumerate(kl)
                if k not in self._index and k in self._host_of]
        if rows:
            idx = torch.tensor([i for i, _ in rows], dtype=torch.int64,
                               device=self.device)
            hs = torch.tensor([h for _, h in rows], dtype=torch.int64)
            out[idx] = self._host_weights[hs].to(self.device)
        return out

    def update_weights(self) -> None:
        super().update_weights()
        self.work_id += 1

    # ---------------------------------------------------------- persistence

    @property
    def num_rows(self) -> int:
        # a key may live in both tiers after persist(); cache is authoritative
        return self._nrows + sum(1 for k in self._host_of
                                 if k not in self._index)

    def should_persist(self) -> bool:
        """Cache filled since the last checkpoint and none pending — time to
        snapshot (reference should_persist, PmemEmbeddingOptimizerVariable.h:84-86
        propagated via EmbeddingPullOperator.cpp:182-189)."""
        return self._cache_full_since_ckpt and self._checkpoint_work_id is None

    def persist(self) -> int:
        """Stamp a checkpoint: flush every dirty cache row to host so the
        host tier holds a complete image, record the watermark (the
        reference flushes items older than the pending checkpoint,
        PmemEmbeddingTable.h:314-328; here one batched flush replaces the
        incremental flush queue)."""
        if self._nrows:
            key_of = {s: k for k, s in self._index.items()}
            w = self.weights[:self._nrows].to("cpu")
            s = self.state[:self._nrows].to("cpu") if self.state_dim else None
            for sl in range(self._nrows):
                hs = self._host_alloc(key_of[sl])
                self._host_weights[hs] = w[sl]
                if s is not None:
                    self._host_state[hs] = s[sl]
        self._checkpoint_work_id = self.work_id
        self._cache_full_since_ckpt = False
        return self.work_id

    def checkpoint_committed(self) -> None:
        """Caller persisted the host image (e.g. wrote it to disk) — clear
        the pending watermark (reference persist_pending_window advance)."""
        self._checkpoint_work_id = None

    # ------------------------------------------------------------ checkpoint

    def export_rows(self, include_state: bool = True):
        """Cache rows + host rows (host rows carry their flushed state)."""
        keys_c, w_c, s_c = super().export_rows(include_state)
        items = sorted(((k, h) for k, h in self._host_of.items()
                        if k not in self._index), key=lambda kv: kv[1])
        if not items:
            return keys_c, w_c, s_c
        hkeys = torch.tensor([k for k, _ in items], dtype=torch.int64,
                             device=self.device)
        hslots = torch.tensor([h for _, h in items], dtype=torch.int64)
        hw = self._host_weights[hslots].to(self.device)
        keys = torch.cat([keys_c, hkeys])
        w = torch.cat([w_c, hw])
        s = None
        if include_state and self.state_dim:
            hs = self._host_state[hslots].to(self.device)
            s = torch.cat([s_c, hs])
        return keys, w, s

    def clear(self) -> None:
        super().clear()
        self._host_of = {}
        self._host_free = []
        self._host_next = 0
        self._touch = torch.zeros(0, dtype=torch.int64)
        self._cache_full_since_ckpt = False
        self._checkpoint_work_id = None
