"""The variable engine: one shard of one embedding variable.

This is the MI355X-native rebuild of the reference's L1/L2 layers
(openembedding/variable/EmbeddingTable.h, EmbeddingOptimizerVariable.h,
server/EmbeddingStorage.h) collapsed into a single per-rank object:

- shard owner of global key k is ``k % shard_num``; local index in array
  mode is ``k // shard_num`` (the reference's layout, EmbeddingShardFile.h:23-25,
  EmbeddingPullOperator.cpp:74-78 — kept for checkpoint compatibility);
- array table (bounded vocabulary) = dense [cap, dim] weights + valid bitmap
  (reference EmbeddingTable.h:171-180 EmbeddingArrayTable);
- hash table (vocabulary >= 2^63, i.e. the full uint64 key space) = open
  addressing key->row-slot over a growable row slab (reference
  EmbeddingHashTable over EmbeddingItemPool);
- rows are created lazily on first pull/update with a deterministic
  initializer (core/rng.py) and optimizer state train_init, matching the
  observable semantics of the reference's ``_new_weights`` side-table +
  commit-time merge (EmbeddingOptimizerVariable.h:242-297);
- gradients pushed for a batch are summed per unique key with occurrence
  counts; ``update_weights`` applies the optimizer once per touched key
  (reference MpscGradientReducer.h:30-53).

The torch backend below runs on CPU (tests, oracle) and on GPU tensors; the
HIP backend (openembedding_amd.ops) replaces the hot paths with fused CDNA4
kernels and is REQUIRED on ROCm devices — on a GPU box a missing extension
raises instead of silently falling back.
"""

from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional, Tuple

import torch

from .initializers import Initializer, make_initializer
from .optimizers import SparseOptimizer, make_optimizer

HASH_VOCAB_THRESHOLD = 1 << 63  # reference Meta.h:44-46 use_hash_table()


@dataclasses.dataclass
class VariableMeta:
    """Wire/JSON metadata (reference openembedding/variable/Meta.h:30-60)."""

    variable_id: int
    embedding_dim: int
    dtype: torch.dtype = torch.float32
    vocabulary_size: int = HASH_VOCAB_THRESHOLD

    @property
    def use_hash_table(self) -> bool:
        return self.vocabulary_size >= HASH_VOCAB_THRESHOLD

    def datatype_str(self) -> str:
        return {torch.float32: "float32", torch.float64: "float64"}[self.dtype]


class VariableShard:
    """One rank's shard of one embedding variable (torch backend)."""

    GROW = 2

    def __init__(self, meta: VariableMeta, shard_id: int = 0, shard_num: int = 1,
                 device: str = "cpu", seed: int = 0):
        self.meta = meta
        self.shard_id = shard_id
        self.shard_num = shard_num
        self.device = torch.device(device)
        self.seed = seed ^ (meta.variable_id * 0x9E3779B97F4A7C15 & (1 << 63) - 1)
        self.dim = meta.embedding_dim
        self.dtype = meta.dtype

        self.optimizer: Optional[SparseOptimizer] = None
        self.initializer: Initializer = make_initializer("constant", value=0.0)
        self.state_dim = 0

        if meta.use_hash_table:
            self._index: Dict[int, int] = {}
            self._array_cap = 0
        else:
            # array table: local capacity covers keys k with k % shard_num == shard_id
            vocab = meta.vocabulary_size
            self._array_cap = (vocab - self.shard_id + self.shard_num - 1) // self.shard_num
            self._index = None
            self.valid = torch.zeros(self._array_cap, dtype=torch.bool,
                                     device=self.device)
            self.slot_key = None  # derived: slot s -> key s*shard_num+shard_id

        cap0 = self._array_cap if not meta.use_hash_table else 0
        self.weights = torch.zeros((cap0, self.dim), dtype=self.dtype,
                                   device=self.device)
        self.state = torch.zeros((cap0, 0), dtype=self.dtype, device=self.device)
        self._nrows = cap0 if not meta.use_hash_table else 0
        self._pending: List[Tuple[torch.Tensor, torch.Tensor, torch.Tensor]] = []
        self._state_init_row: Optional[torch.Tensor] = None

    # ------------------------------------------------------------------ config

    def set_initializer(self, category: str, **cfg) -> None:
        self.initializer = make_initializer(category, **cfg)

    def set_optimizer(self, category: str, **cfg) -> None:
        """Install/replace the optimizer, keeping weights; state is rebuilt
        (reference live-reconfig semantics, EmbeddingVariable.cpp:29-60:
        state is migrated only when the optimizer category is unchanged)."""
        new_opt = make_optimizer(category, **cfg)
        keep_state = (self.optimizer is not None
                      and self.optimizer.category == category
                      and new_opt.state_dim(self.dim) == self.state_dim)
        self.optimizer = new_opt
        if not keep_state:
            self.state_dim = new_opt.state_dim(self.dim)
            self.state = torch.zeros((self.weights.shape[0], self.state_dim),
                                     dtype=self.dtype, device=self.device)
            if self.state_dim:
                # train_init state for every existing row
                if self.meta.use_hash_table:
                    if self._nrows:
                        new_opt.train_init(self.state[:self._nrows], self.dim)
                else:
                    if bool(self.valid.any()):
                        sl = self.valid.nonzero(as_tuple=True)[0]
                        s = self.state[sl]
                        new_opt.train_init(s, self.dim)
                        self.state[sl] = s
        self._state_init_row = self._make_state_init_row()

    def _make_state_init_row(self) -> torch.Tensor:
        row = torch.zeros((1, self.state_dim), dtype=self.dtype, device=self.device)
        if self.optimizer is not None and self.state_dim:
            self.optimizer.train_init(row, self.dim)
        return row

    # ------------------------------------------------------------- row storage

    def _ensure_rows(self, need: int) -> None:
        cap = self.weights.shape[0]
        if need <= cap:
            return
        new_cap = max(need, max(1024, cap * self.GROW))
        w = torch.zeros((new_cap, self.dim), dtype=self.dtype, device=self.device)
        w[:cap] = self.weights
        self.weights = w
        s = torch.zeros((new_cap, self.state_dim), dtype=self.dtype,
                        device=self.device)
        s[:cap] = self.state
        self.state = s

    def _lookup_or_insert(self, keys: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """keys: unique int64 global keys owned by this shard.
        Returns (slots int64 [n], new_mask bool [n]); creates rows for new keys
        (weights via initializer, state via train_init)."""
        if self.meta.use_hash_table:
            slots = torch.empty(keys.numel(), dtype=torch.int64)
            new_mask = torch.zeros(keys.numel(), dtype=torch.bool)
            idx = self._index
            nxt = self._nrows
            for i, k in enumerate(keys.tolist()):
                if k == -1:
                    # reserved: the hash table's empty marker — the
                    # reference reserves the same value (empty_key = -1,
                    # EmbeddingVariable.cpp:21). The GPU backend resolves
                    # it to a zero row; the CPU oracle fails loudly so the
                    # mistake surfaces in development.
                    raise ValueError(
                        "key -1 is reserved in hash mode (table empty "
                        "marker; same reservation as the reference)")
                s = idx.get(k)
                if s is None:
                    s = nxt
                    idx[k] = s
                    nxt += 1
                    new_mask[i] = True
                slots[i] = s
            slots = slots.to(self.device)
            new_mask = new_mask.to(self.device)
            if nxt != self._nrows:
                self._ensure_rows(nxt)
                self._nrows = nxt
        else:
            slots = keys // self.shard_num
            if keys.numel():
                bad = (keys % self.shard_num != self.shard_id)
                if bool(bad.any()):
                    raise ValueError("keys not owned by this shard")
                if bool((slots >= self._array_cap).any()) or bool((slots < 0).any()):
                    raise IndexError("key out of vocabulary range")
            new_mask = ~self.valid[slots]
            self.valid[slots] = True
        if bool(new_mask.any()):
            nk = keys[new_mask]
            ns = slots[new_mask]
            self.weights[ns] = self.initializer(self.seed, nk, self.dim, self.dtype)
            if self.state_dim:
                self.state[ns] = self._state_init_row.expand(ns.numel(), -1)
        return slots, new_mask

    def _lookup_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        """Slots for existing keys; -1 for missing (read-only/serving path)."""
        if self.meta.use_hash_table:
            idx = self._index
            return torch.tensor([idx.get(k, -1) for k in keys.tolist()],
                                dtype=torch.int64, device=self.device)
        slots = keys // self.shard_num
        slots = torch.where(self.valid[slots.clamp(0, self._array_cap - 1)],
                            slots, torch.full_like(slots, -1))
        return slots

    # ---------------------------------------------------------------- training

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        """keys: unique int64 [n] owned by this shard -> weights [n, dim].
        Missing rows are created and initialized (reference
        EmbeddingOptimizerVariable.h:242-266)."""
        slots, _ = self._lookup_or_insert(keys)
        return self.weights[slots].clone()

    def pull_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        """Serving-mode pull: missing rows come back zero, table untouched
        (reference read_only get_weights path, EmbeddingPullOperator.cpp:179-181)."""
        slots = self._lookup_readonly(keys)
        out = torch.zeros((keys.numel(), self.dim), dtype=self.dtype,
                          device=self.device)
        hit = slots >= 0
        out[hit] = self.weights[slots[hit]]
        return out

    def push(self, keys: torch.Tensor, grads: torch.Tensor,
             counts: torch.Tensor) -> None:
        """Queue a pre-aggregated gradient block (keys unique within block,
        grads summed, counts = occurrence counts). Reference
        MpscGradientReducer.h:26-29 push side."""
        self._pending.append((keys, grads, counts))

    def update_weights(self) -> None:
        """Commit the batch: merge pending blocks by key, apply the optimizer
        once per unique key (reference EmbeddingOptimizerVariable.h:273-297)."""
        if not self._pending:
            return
        if self.optimizer is None:
            raise RuntimeError("update_weights called before set_optimizer")
        if len(self._pending) == 1:
            keys, grads, counts = self._pending[0]
        else:
            keys = torch.cat([b[0] for b in self._pending])
            grads = torch.cat([b[1] for b in self._pending])
            counts = torch.cat([b[2] for b in self._pending])
            uk, inv = torch.unique(keys, return_inverse=True)
            g = torch.zeros((uk.numel(), self.dim), dtype=grads.dtype,
                            device=self.device)
            g.index_add_(0, inv, grads)
            c = torch.zeros(uk.numel(), dtype=counts.dtype, device=self.device)
            c.index_add_(0, inv, counts)
            keys, grads, counts = uk, g, c
        self._pending = []
        slots, _ = self._lookup_or_insert(keys)
        w = self.weights[slots]
        s = self.state[slots]
        self.optimizer.update(w, s, counts, grads)
        self.weights[slots] = w
        self.state[slots] = s

    # ------------------------------------------------------------- checkpoint

    @property
    def num_rows(self) -> int:
        if self.meta.use_hash_table:
            return self._nrows
        return int(self.valid.sum())

    def export_rows(self, include_state: bool = True
                    ) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
        """All materialized rows: (keys [n], weights [n,dim], state [n,sd] or None).
        Keys come back in stable (slot) order."""
        if self.meta.use_hash_table:
            items = sorted(self._index.items(), key=lambda kv: kv[1])
            keys = torch.tensor([k for k, _ in items], dtype=torch.int64,
                                device=self.device)
            slots = torch.tensor([s for _, s in items], dtype=torch.int64,
                                 device=self.device)
        else:
            slots = self.valid.nonzero(as_tuple=True)[0]
            keys = slots * self.shard_num + self.shard_id
        w = self.weights[slots].clone()
        s = self.state[slots].clone() if (include_state and self.state_dim) else None
        return keys, w, s

    def import_rows(self, keys: torch.Tensor, weights: torch.Tensor,
                    state: Optional[torch.Tensor] = None) -> None:
        """Bulk set rows (load path). State==None leaves state at train_init
        (reference load with include_optimizer=false)."""
        slots, _ = self._lookup_or_insert(keys)
        self.weights[slots] = weights.to(self.device, self.dtype)
        if state is not None and self.state_dim:
            self.state[slots] = state.to(self.device, self.dtype)

    def get_weights(self, keys: torch.Tensor) -> torch.Tensor:
        return self.pull_readonly(keys)

    def clear(self) -> None:
        """Drop all rows (reference clear_weights, used before load)."""
        if self.meta.use_hash_table:
            self._index = {}
            self._nrows = 0
            self.weights = torch.zeros((0, self.dim), dtype=self.dtype,
                                       device=self.device)
            self.state = torch.zeros((0, self.state_dim), dtype=self.dtype,
                                     device=self.device)
        else:
            self.valid.zero_()
            self.weights.zero_()
            self.state.zero_()
