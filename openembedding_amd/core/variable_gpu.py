"""GPU (ROCm/MI355X) backend of the variable shard.

Same contract as core.variable.VariableShard, with the hot paths running the
in-tree CDNA4 kernels (ops/csrc/embops.hip):

  - hash mode: open-addressed key->slot table in HBM (splitmix64 probe),
    bump-allocated row slab; capacity management is HOST-side and sync-free
    on the pull path (a conservative upper bound of live rows is maintained
    and the exact device counter is read once per committed batch);
  - array mode: dense rows + valid bitmap, preallocated for the shard's
    vocabulary slice (288 GB HBM per GPU makes full preallocation the right
    trade — no grow path on the hot path);
  - gather+lazy-init, reduce-by-key and the fused optimizer run as single
    kernels on the current stream.

The extension is REQUIRED here — construction raises if it is missing
(no silent eager fallback on a GPU box).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..ops import require_hip
from .variable import VariableMeta, VariableShard

_OPT_IDS = {"default": 0, "adadelta": 1, "adagrad": 2, "adam": 3,
            "adamax": 4, "ftrl": 5, "rmsprop": 6, "sgd": 7, "test": 8}

_INIT_IDS = {"constant": 0, "uniform": 1, "normal": 2}


def _opt_cfg_vector(opt) -> list:
    c = opt.cfg
    cat = opt.category
    if cat == "default":
        return [c["learning_rate"]]
    if cat == "adadelta":
        return [c["learning_rate"], c["rho"], c["epsilon"]]
    if cat == "adagrad":
        return [c["learning_rate"], c["initial_accumulator_value"], c["epsilon"]]
    if cat == "adam" or cat == "adamax":
        return [c["learning_rate"], c["beta_1"], c["beta_2"], c["epsilon"]]
    if cat == "ftrl":
        return [c["learning_rate"], c["l1_regularization_strength"],
                c["l2_regularization_strength"],
                c["l2_shrinkage_regularization_strength"],
                c["learning_rate_power"], c["beta"]]
    if cat == "rmsprop":
        return [c["learning_rate"], c["rho"], c["momentum"], c["epsilon"]]
    if cat == "sgd":
        return [c["learning_rate"], c["momentum"], 1.0 if c["nesterov"] else 0.0]
    if cat == "test":
        return [c["learning_rate"], c["flip"]]
    raise ValueError(cat)


def _init_params(init) -> Tuple[int, float, float, float]:
    cat = _INIT_IDS[init.category]
    c = init.cfg
    if init.category == "constant":
        return cat, c["value"], 0.0, 0.0
    if init.category == "uniform":
        return cat, c["minval"], c["maxval"], 0.0
    return cat, c["mean"], c["stddev"], c["truncated"]


class HipVariableShard(VariableShard):
    """VariableShard with HIP-kernel hot paths (float32 rows)."""

    INITIAL_TABLE_CAP = 1 << 16
    INITIAL_ROW_CAP = 1 << 14

    def __init__(self, meta: VariableMeta, shard_id: int = 0, shard_num: int = 1,
                 device: str = "cuda", seed: int = 0):
        self.ext = require_hip()
        if self.ext is None:
            raise RuntimeError("HIP extension missing (OEAMD_ALLOW_TORCH_FALLBACK "
                               "cannot be used with HipVariableShard)")
        if meta.dtype != torch.float32:
            raise NotImplementedError("GPU shards are float32 (reference default); "
                                      "float64 runs on the CPU backend")
        super().__init__(meta, shard_id, shard_num, device, seed)
        self._pending_bounded = []
        dev = self.device
        if meta.use_hash_table:
            self._cap = self.INITIAL_TABLE_CAP
            self.tk = torch.full((self._cap,), -1, dtype=torch.int64, device=dev)
            self.tv = torch.empty(self._cap, dtype=torch.int32, device=dev)
            self.nrows_dev = torch.zeros(1, dtype=torch.int32, device=dev)
            self.slot_keys = torch.empty(self.INITIAL_ROW_CAP, dtype=torch.int64,
                                         device=dev)
            self.weights = torch.zeros((self.INITIAL_ROW_CAP, self.dim),
                                       dtype=self.dtype, device=dev)
            self.state = torch.zeros((self.INITIAL_ROW_CAP, 0),
                                     dtype=self.dtype, device=dev)
            self._nrows_upper = 0   # conservative (>= true count)
            self._nrows_exact = 0   # refreshed once per committed batch
        else:
            self.valid_u8 = torch.zeros(self._array_cap, dtype=torch.uint8,
                                        device=dev)
            del self.valid  # base bool bitmap unused on GPU

    # ------------------------------------------------------------- capacity

    def _ensure_rows(self, need: int) -> None:
        cap = self.weights.shape[0]
        if need <= cap:
            return
        new_cap = max(need, cap * 2)
        for name in ("weights", "state"):
            old = getattr(self, name)
            neww = torch.zeros((new_cap, old.shape[1]), dtype=self.dtype,
                               device=self.device)
            neww[:cap] = old
            setattr(self, name, neww)
        sk = torch.empty(new_cap, dtype=torch.int64, device=self.device)
        sk[:self.slot_keys.shape[0]] = self.slot_keys
        self.slot_keys = sk

    def _maybe_rehash(self) -> None:
        if self._nrows_upper * 2 <= self._cap:
            return
        new_cap = self._cap
        while self._nrows_upper * 2 > new_cap:
            new_cap *= 2
        tk_new = torch.empty(new_cap, dtype=torch.int64, device=self.device)
        tv_new = torch.empty(new_cap, dtype=torch.int32, device=self.device)
        self.ext.ht_rehash(self.tk, self.tv, tk_new, tv_new)
        self.tk, self.tv, self._cap = tk_new, tv_new, new_cap

    # -------------------------------------------------------------- lookups

    def reserve_rows(self, rows: int) -> None:
        """Pre-size the row slab and probe table for ``rows`` keys and mark
        the shard capture-safe: with capacity fixed up front, the insert
        path has no host-side growth and a hipGraph may capture it (the
        kernel's capacity guard turns overflow beyond the reservation into
        dropped rows instead of corruption). CombinedEmbedding in hash mode
        auto-reserves its key-space size."""
        rows = int(rows)
        self._ensure_rows(max(rows, 1024))
        want_cap = 1 << max(16, (2 * rows - 1).bit_length())
        if want_cap > self._cap:
            old_upper = self._nrows_upper
            self._nrows_upper = want_cap // 2
            self._maybe_rehash()
            self._nrows_upper = old_upper
        self._reserved_rows = rows

    def _lookup_or_insert(self, keys: torch.Tensor, u_dev=None):
        n = keys.numel()
        if self.meta.use_hash_table:
            if self._in_graph_capture():
                if getattr(self, "_reserved_rows", 0) <= 0:
                    # growth (slab realloc, rehash) needs host control; a
                    # captured insert would write past the frozen capacity
                    # on replay. Raising makes graph capture fall back to
                    # eager (bench.py catches it). The kernel additionally
                    # bounds the slot counter by the slab capacity.
                    raise RuntimeError(
                        "hash-table insert path is not hipGraph-capturable "
                        "without reserve_rows()")
                return self.ext.ht_lookup(self.tk, self.tv, keys,
                                          self.nrows_dev, self.slot_keys,
                                          True, u_dev)
            self._ensure_rows(self._nrows_upper + n)
            self._maybe_rehash()
            slots, new_mask = self.ext.ht_lookup(self.tk, self.tv, keys,
                                                 self.nrows_dev, self.slot_keys,
                                                 True, u_dev)
            self._nrows_upper += n
        else:
            slots, new_mask = self.ext.array_touch(self.valid_u8, keys,
                                                   self.shard_num, u_dev)
        return slots, new_mask

    def _lookup_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        if self.meta.use_hash_table:
            slots, _ = self.ext.ht_lookup(self.tk, self.tv, keys,
                                          self.nrows_dev, self.slot_keys,
                                          False, None)
            return slots
        slots = torch.div(keys, self.shard_num, rounding_mode="floor")
        hit = self.valid_u8[slots.clamp(0, self._array_cap - 1)] != 0
        return torch.where(hit, slots, torch.full_like(slots, -1))

    # ------------------------------------------------------------- hot paths

    def _gather(self, keys, slots, new_mask, want_out: bool,
                inverse=None, u_dev=None) -> torch.Tensor:
        cat, p0, p1, p2 = _init_params(self.initializer)
        sir = self._state_init_row
        if sir is None:
            sir = self._make_state_init_row()
            self._state_init_row = sir
        return self.ext.gather_init(self.weights, self.state, slots,
                                    new_mask if new_mask is not None
                                    else torch.empty(0, dtype=torch.uint8,
                                                     device=self.device),
                                    keys, inverse, cat, p0, p1, p2, self.seed,
                                    sir.reshape(-1), want_out, u_dev)

    def pull(self, keys: torch.Tensor) -> torch.Tensor:
        slots, new_mask = self._lookup_or_insert(keys)
        return self._gather(keys, slots, new_mask, True)

    def pull_bounded(self, keys_buf: torch.Tensor, u_dev: torch.Tensor,
                     inverse: torch.Tensor):
        """Sync-free pull: keys_buf is an n-sized unique buffer with the live
        count in u_dev (device int32); the gather iterates the full inverse
        and fuses the duplicate scatter. Returns (out [n_elems, dim], slots)."""
        slots, new_mask = self._lookup_or_insert(keys_buf, u_dev)
        out = self._gather(keys_buf, slots, new_mask, True,
                           inverse=inverse, u_dev=u_dev)
        return out, slots

    def push_slots(self, keys_buf, u_dev, slots, grads, counts) -> None:
        """Queue a bounded pre-reduced block whose table slots are already
        known (saved from the matching pull)."""
        self._pending_bounded.append((keys_buf, u_dev, slots, grads, counts))

    def pull_readonly(self, keys: torch.Tensor) -> torch.Tensor:
        slots = self._lookup_readonly(keys)
        empty = torch.empty(0, dtype=torch.uint8, device=self.device)
        return self.ext.gather_init(self.weights, self.state, slots, empty,
                                    keys, None, 0, 0.0, 0.0, 0.0, self.seed,
                                    torch.empty(0, dtype=self.dtype,
                                                device=self.device), True,
                                    None)

    def update_weights(self) -> None:
        if not self._pending and not self._pending_bounded:
            return
        if self.optimizer is None:
            raise RuntimeError("update_weights called before set_optimizer")
        opt_id = _OPT_IDS[self.optimizer.category]
        cfg = _opt_cfg_vector(self.optimizer)
        blocks = self._pending_bounded
        self._pending_bounded = []
        if len(blocks) > 1:
            # several pulls committed together: reference semantics
            # (MpscGradientReducer.h:30-53) are ONE optimizer step per
            # unique key over the summed grads+counts — merge the blocks
            # (sync-free: tail-mask, concat, re-dedup) instead of applying
            # them sequentially
            blocks = [self._merge_bounded(blocks)]
        # bounded blocks: slots known from pull, zero-sync apply
        for keys_buf, u_dev, slots, grads, counts in blocks:
            self.ext.apply_optimizer(opt_id, self.weights, self.state, slots,
                                     grads.contiguous(), counts, cfg, u_dev)
        if self._pending:
            if len(self._pending) == 1:
                keys, grads, counts = self._pending[0]
            else:
                keys0 = torch.cat([b[0] for b in self._pending])
                grads0 = torch.cat([b[1] for b in self._pending])
                counts0 = torch.cat([b[2] for b in self._pending])
                keys, inv = self.ext.unique_inverse(keys0)
                grads, _ = self.ext.reduce_by_inverse(inv, grads0, keys.numel())
                counts = torch.zeros(keys.numel(), dtype=counts0.dtype,
                                     device=self.device)
                counts.index_add_(0, inv, counts0)
            self._pending = []
            slots, new_mask = self._lookup_or_insert(keys)
            # init rows that were pushed without a prior pull
            self._gather(keys, slots, new_mask, False)
            self.ext.apply_optimizer(opt_id, self.weights, self.state, slots,
                                     grads.contiguous(), counts, cfg, None)
        if self.meta.use_hash_table and not self._in_graph_capture():
            # every 16 commits: tighten the row-count bound (the D2H read
            # syncs the stream — a per-step sync costs a pipeline bubble;
            # between tightenings the bound only inflates by batch size)
            self._commits = getattr(self, "_commits", 0) + 1
            if self._commits % 16 == 0 or self._nrows_upper > (
                    self.weights.shape[0] * 3) // 4:
                self._nrows_exact = int(self.nrows_dev.item())
                self._nrows_upper = self._nrows_exact

    def _merge_bounded(self, blocks):
        """Merge several bounded gradient blocks into one: mask each
        block's garbage tail with the reserved key -1 / zero payload, then
        concatenate and re-dedup. -1 entries dedup into their own uids with
        slot -1 and are skipped by the apply kernel. Zero host syncs."""
        ks, ps = [], []
        for keys_buf, u_dev, slots, grads, counts in blocks:
            self.ext.mask_tail(keys_buf, grads, counts, u_dev)
            ks.append(keys_buf)
            ps.append(torch.cat([grads,
                                 counts.to(grads.dtype).unsqueeze(1)],
                                dim=1))
        keys = torch.cat(ks)
        payload = torch.cat(ps)
        uk, inv, u_dev = self.ext.unique_bounded(keys)
        g2c, _ = self.ext.reduce_by_inverse(inv, payload.contiguous(),
                                            keys.numel())
        g2, c2 = self.ext.split_payload(g2c, u_dev)
        slots, new_mask = self._lookup_or_insert(uk, u_dev)
        # init rows that were pushed without a prior pull
        self._gather(uk, slots, new_mask, False, u_dev=u_dev)
        return uk, u_dev, slots, g2, c2

    @staticmethod
    def _in_graph_capture() -> bool:
        try:
            return torch.cuda.is_current_stream_capturing()
        except Exception:
            return False

    # ----------------------------------------------------------- checkpoint

    @property
    def num_rows(self) -> int:
        if self.meta.use_hash_table:
            return int(self.nrows_dev.item())
        return int((self.valid_u8 != 0).sum())

    def export_rows(self, include_state: bool = True):
        if self.meta.use_hash_table:
            # cached-row count, NOT self.num_rows: the tier subclass
            # overrides num_rows to include host-resident rows, which are
            # exported by ITS export_rows — using it here would read stale
            # slab entries beyond the live count
            n = int(self.nrows_dev.item())
            keys = self.slot_keys[:n].clone()
            w = self.weights[:n].clone()
            s = self.state[:n].clone() if (include_state and self.state_dim) else None
            return keys, w, s
        slots = (self.valid_u8 != 0).nonzero(as_tuple=True)[0]
        keys = slots * self.shard_num + self.shard_id
        w = self.weights[slots].clone()
        s = self.state[slots].clone() if (include_state and self.state_dim) else None
        return keys, w, s

    def import_rows(self, keys, weights, state=None):
        slots, new_mask = self._lookup_or_insert(keys)
        self._gather(keys, slots, new_mask, False)  # init state for new rows
        self.weights[slots] = weights.to(self.device, self.dtype)
        if state is not None and self.state_dim:
            self.state[slots] = state.to(self.device, self.dtype)
        if self.meta.use_hash_table:
            self._nrows_exact = int(self.nrows_dev.item())
            self._nrows_upper = self._nrows_exact

    def get_weights(self, keys):
        return self.pull_readonly(keys)

    def clear(self) -> None:
        if self.meta.use_hash_table:
            self.tk.fill_(-1)
            self.nrows_dev.zero_()
            self._nrows_upper = 0
            self._nrows_exact = 0
        else:
            self.valid_u8.zero_()
            self.weights.zero_()
            self.state.zero_()

    # state handling shared with base -------------------------------------

    def set_optimizer(self, category: str, **cfg) -> None:
        # base logic works (torch tensor ops on GPU); sync the fields it reads
        if self.meta.use_hash_table:
            self._nrows = self.num_rows
            super().set_optimizer(category, **cfg)
        else:
            self.valid = self.valid_u8 != 0  # view for base set_optimizer
            super().set_optimizer(category, **cfg)
            del self.valid
