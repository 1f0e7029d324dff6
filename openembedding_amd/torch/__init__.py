"""openembedding_amd.torch — the user-facing module API.

PyTorch rebuild of the reference's Keras surface
(reference openembedding/tensorflow/exb.py): ``Embedding``,
``distributed_optimizer``, ``distributed_model``, ``Model``,
``save_server_model`` / ``load_server_model``, ``save_as_original_model``,
``pulling``. The fake-gradient + registered-gradient machinery of the
reference (exb.py:89-104) collapses into one autograd.Function whose forward
is the sharded pull and whose backward is the sharded push.

Typical use (3-line change, like the reference README):

    import openembedding_amd.torch as embed
    model.embedding = embed.Embedding(-1, 64)          # PS-backed, hashed keys
    opt = embed.distributed_optimizer(torch.optim.Adagrad(model.parameters(), lr=0.01))
    ...
    loss.backward(); opt.step()                        # commits sparse + dense
"""

from __future__ import annotations

import math
import weakref
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist
import torch.nn as nn

from ..context import Context, Storage, get_context
from ..core.variable import HASH_VOCAB_THRESHOLD
from ..parallel import comm
from ..parallel.sharded import ShardedVariable

__all__ = [
    "Embedding", "CombinedEmbedding", "Variable", "Context",
    "distributed_optimizer", "DistributedOptimizer", "distributed_model",
    "Model", "save_server_model", "load_server_model",
    "save_as_original_model", "pulling", "sparse_read_as_dense",
    "should_persist_server_model", "persist_server_model",
    "restore_server_model", "get_context",
    "Adadelta", "Adagrad", "Adam", "Adamax", "Ftrl", "Nadam", "NAdam",
    "RMSprop", "SGD",
]

# registry of live PS-backed embeddings (reference track_variable,
# exb.py:125-131); weakrefs so dropped models don't pin their engine state
_tracked: List = []


def _tracked_live():
    out = []
    dead = False
    for r in _tracked:
        e = r() if isinstance(r, weakref.ref) else r
        if e is None:
            dead = True
        else:
            out.append(e)
    if dead:
        _tracked[:] = [r for r in _tracked
                       if not (isinstance(r, weakref.ref) and r() is None)]
    return out


class _PullPushFn(torch.autograd.Function):
    """forward = sharded pull, backward = sharded push (replaces the
    reference's PullWeights op + fake-grad PushGradients registration)."""

    @staticmethod
    def forward(ctx, hook: torch.Tensor, indices: torch.Tensor,
                var: ShardedVariable, pre=None, field_offsets=None):
        if pre is None:
            out, handle = var.pull(indices, field_offsets=field_offsets)
        else:  # prefetched (out, handle) from Variable.prefetch
            out, handle = pre
        ctx.var = var
        ctx.handle = handle
        return out

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        # cast to the VARIABLE's dtype (f32 normally; f64 tables keep f64)
        ctx.var.push(ctx.handle,
                     grad_out.contiguous().to(ctx.var.shard.dtype))
        return (torch.zeros(0, device=grad_out.device), None, None, None,
                None)


class Variable:
    """Thin user handle over a ShardedVariable (reference exb.py Variable,
    :222-386): sparse_read + prefetch + explicit push/update."""

    def __init__(self, sharded: ShardedVariable, storage: Storage):
        self.sharded = sharded
        self.storage = storage
        self._prefetched: List = []  # FIFO of (match_id, out, handle, event)

    def prefetch(self, indices: torch.Tensor, match_ref=None) -> None:
        """Issue the pull for a FUTURE batch now, on the prefetch stream
        (reference PrefetchPullWeights issued from the dataset thread,
        exb_ops.cpp:109-205). ``match_ref`` identifies the batch at consume
        time (default: the indices tensor itself). Collective —
        every rank must prefetch the same variables in the same order."""
        ctx = get_context()
        if match_ref is None:
            match_ref = indices
        match_id = id(match_ref)
        if ctx.device.type == "cuda":
            stream = ctx.prefetch_stream
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                out, handle = self.sharded.pull(indices)
                ev = torch.cuda.Event()
                ev.record(stream)
        else:
            out, handle = self.sharded.pull(indices)
            ev = None
        # the queue holds a strong reference to the match object: without
        # it CPython could reuse the id() of a dropped batch and a later
        # unrelated tensor would consume the stale (out, handle) pair
        self._prefetched.append((match_id, out, handle, ev, match_ref))

    def _take_prefetched(self, indices: torch.Tensor):
        """Pop the prefetched entry for these indices, or None. FIFO with
        identity check (the reference validated with a sampled-index hash,
        Prefetch.h:34-44; object identity is exact here because pulling()
        passes the same tensors through)."""
        want = id(indices)
        for pos, (match_id, out, handle, ev, _ref) in enumerate(
                self._prefetched):
            if match_id == want:
                # drop skipped older entries along with the match
                del self._prefetched[:pos + 1]
                if ev is not None:
                    torch.cuda.current_stream().wait_event(ev)
                return out, handle
        if len(self._prefetched) > 8:  # never-consumed entries: bound memory
            self._prefetched.pop(0)
        return None

    @property
    def embedding_dim(self):
        return self.sharded.embedding_dim

    @property
    def vocabulary_size(self):
        return self.sharded.meta.vocabulary_size

    def sparse_read(self, indices: torch.Tensor) -> torch.Tensor:
        out, _ = self.sharded.pull(indices, readonly=True)
        return out

    def set_initializer(self, category: str, **cfg):
        self.sharded.set_initializer(category, **cfg)

    def set_optimizer(self, category: str, **cfg):
        self.sharded.set_optimizer(category, **cfg)

    def update_weights(self):
        self.sharded.update_weights()


class Embedding(nn.Module):
    """PS-backed embedding layer (reference exb.Embedding, exb.py:388-443).

    num_embeddings == -1 selects hash mode over the full int64 key space
    (reference input_dim=-1 -> 2^63, exb.py:393-396); otherwise a bounded
    array table sharded across ranks.

    ``sparse_as_dense``: keep this table as an ordinary replicated
    nn.Embedding trained by allreduce — the reference's "cache" policy for
    small hot vocabularies (exb.py:241-248, ~+10%: documents/en/benchmark.md).
    """

    def __init__(self, num_embeddings: int, embedding_dim: int,
                 initializer: Optional[Dict] = None,
                 num_shards: Optional[int] = None,
                 sparse_as_dense: bool = False,
                 storage: Optional[Storage] = None,
                 dtype: torch.dtype = torch.float32):
        # num_shards is accepted for API parity (reference exb.py:276
        # decoupled shard count from server count); here shards == ranks by
        # construction and checkpoints reload across any world size, so the
        # value is advisory only.
        super().__init__()
        ctx = get_context()
        self.ctx = ctx
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.sparse_as_dense = sparse_as_dense
        if sparse_as_dense:
            if num_embeddings < 0:
                raise ValueError("sparse_as_dense needs a bounded vocabulary")
            self.dense = nn.Embedding(num_embeddings, embedding_dim,
                                      dtype=dtype)
            self.dense.to(ctx.device)
            self.variable = None
        else:
            vocab = HASH_VOCAB_THRESHOLD if num_embeddings < 0 else num_embeddings
            st = storage if storage is not None else _default_storage(ctx)
            sharded = st.create_variable(vocab, embedding_dim, dtype)
            init = initializer or {"category": "uniform",
                                   "minval": -1.0 / max(1, int(math.sqrt(embedding_dim))),
                                   "maxval": 1.0 / max(1, int(math.sqrt(embedding_dim)))}
            cat = init.pop("category") if "category" in init else "uniform"
            sharded.set_initializer(cat, **init)
            self.variable = Variable(sharded, st)
            # zero-size hook so autograd reaches our Function even though
            # indices carry no grad
            self.grad_hook = nn.Parameter(torch.zeros(0, device=ctx.device))
            _tracked.append(weakref.ref(self))

    def forward(self, indices: torch.Tensor) -> torch.Tensor:
        return self._forward_keys(indices, indices)

    def _forward_keys(self, keys: torch.Tensor,
                      match_ref: torch.Tensor,
                      field_offsets: torch.Tensor = None) -> torch.Tensor:
        # field_offsets: keys are RAW per-field ids; the GPU pull fuses the
        # offset add into its unique kernels (CombinedEmbedding hot path)
        if self.sparse_as_dense:
            if field_offsets is not None:
                keys = keys + field_offsets
            return self.dense(keys)
        if torch.is_grad_enabled() and self.grad_hook.requires_grad:
            pre = self.variable._take_prefetched(match_ref)
            return _PullPushFn.apply(self.grad_hook, keys,
                                     self.variable.sharded, pre,
                                     field_offsets)
        if field_offsets is not None:
            keys = keys + field_offsets
        return self.variable.sparse_read(keys)

    def prefetch(self, indices: torch.Tensor) -> None:
        """Issue this batch's pull ahead of time (reference prefetch,
        exb.py:331-340)."""
        if not self.sparse_as_dense:
            self.variable.prefetch(indices)

    def extra_repr(self):
        mode = ("dense" if self.sparse_as_dense else
                ("hash" if self.num_embeddings < 0 else "array"))
        return f"{self.num_embeddings}, {self.embedding_dim}, mode={mode}"


class CombinedEmbedding(Embedding):
    """Many categorical fields sharing one PS variable.

    MI355X-first design: instead of the reference's one-Embedding-per-feature (26
    pulls -> 26 RPC fan-outs per step for Criteo), all fields share one key
    space with per-field offsets, so a step does ONE unique+all_to_all+gather
    for all fields. forward takes [batch, n_fields] raw per-field ids and
    returns [batch, n_fields, dim].
    """

    def __init__(self, field_vocab_sizes: Sequence[int], embedding_dim: int,
                 hash_mode: bool = False, **kw):
        sizes = list(field_vocab_sizes)
        offsets = [0]
        for s in sizes:
            offsets.append(offsets[-1] + int(s))
        # hash_mode: same offset key space, but stored in the open-addressed
        # hash table (rows created lazily; enables the capacity tier)
        super().__init__(-1 if hash_mode else offsets[-1], embedding_dim,
                         **kw)
        if hash_mode and self.variable is not None:
            # the key space is known (field offsets): pre-size the table so
            # the GPU insert path stays hipGraph-capturable
            self.variable.sharded.reserve_rows(offsets[-1])
        self.n_fields = len(sizes)
        self.register_buffer(
            "field_offsets",
            torch.tensor(offsets[:-1], dtype=torch.int64,
                         device=get_context().device))

    def forward(self, field_ids: torch.Tensor) -> torch.Tensor:
        ids = field_ids.to(torch.int64)
        if ids.is_cuda and not self.sparse_as_dense:
            # raw ids through; the GPU pull fuses `+ field_offsets` into
            # its unique kernels (saves the broadcast-add launch + the
            # [B, F] int64 intermediate every step). Prefetch matching is
            # by the ORIGINAL field_ids tensor either way.
            return self._forward_keys(ids, field_ids,
                                      field_offsets=self.field_offsets)
        keys = ids + self.field_offsets
        return self._forward_keys(keys, field_ids)

    def prefetch(self, field_ids: torch.Tensor) -> None:
        if self.sparse_as_dense:
            return
        keys = field_ids.to(torch.int64) + self.field_offsets
        self.variable.prefetch(keys, match_ref=field_ids)


def _default_storage(ctx: Context) -> Storage:
    if not ctx.storages:
        ctx.create_storage()
    return ctx.storages[0]


# --------------------------------------------------------------- optimizer

_TORCH_TO_SERVER = {
    "SGD": "sgd", "Adagrad": "adagrad", "Adam": "adam", "AdamW": "adam",
    "Adamax": "adamax", "Adadelta": "adadelta", "RMSprop": "rmsprop",
}


def _server_cfg_from_torch(opt: torch.optim.Optimizer) -> Dict:
    """Translate a torch optimizer's hyper-params to the server-side sparse
    optimizer config (the reference's TF->category translation,
    exb.py:64-86)."""
    name = type(opt).__name__
    cat = _TORCH_TO_SERVER.get(name)
    if cat is None:
        raise ValueError(f"no server-side sparse optimizer for {name}; "
                         f"set one explicitly with Embedding.variable."
                         f"set_optimizer(...)")
    g = opt.param_groups[0]
    lr = g["lr"]
    if cat == "sgd":
        return dict(category="sgd", learning_rate=lr,
                    momentum=g.get("momentum", 0.0),
                    nesterov=bool(g.get("nesterov", False)))
    if cat == "adagrad":
        return dict(category="adagrad", learning_rate=lr,
                    initial_accumulator_value=g.get("initial_accumulator_value", 0.0),
                    epsilon=g.get("eps", 1e-10))
    if cat == "adam":
        b1, b2 = g.get("betas", (0.9, 0.999))
        return dict(category="adam", learning_rate=lr, beta_1=b1, beta_2=b2,
                    epsilon=g.get("eps", 1e-8))
    if cat == "adamax":
        b1, b2 = g.get("betas", (0.9, 0.999))
        return dict(category="adamax", learning_rate=lr, beta_1=b1, beta_2=b2,
                    epsilon=g.get("eps", 1e-8))
    if cat == "adadelta":
        return dict(category="adadelta", learning_rate=lr,
                    rho=g.get("rho", 0.9), epsilon=g.get("eps", 1e-6))
    if cat == "rmsprop":
        return dict(category="rmsprop", learning_rate=lr,
                    rho=g.get("alpha", 0.99), momentum=g.get("momentum", 0.0),
                    epsilon=g.get("eps", 1e-8))
    raise AssertionError(cat)


class _FlatDenseOptimizer:
    """Dense params re-based onto ONE flat buffer: zero_grad is a no-op in
    steady state (the step kernel re-zeroes the grad buffer it consumes),
    the distributed allreduce is one collective on one tensor, and the
    optimizer step is a handful of flat kernels (torch's multi-tensor
    apply was ~60us + ~60us of per-param grad fills per step on the DeepFM
    profile). Supports plain Adagrad, SGD (momentum/nesterov) and Adam —
    numerically identical to the torch.optim defaults (no weight decay /
    lr_decay / amsgrad); other optimizers keep the non-flat path."""

    _FLAT_IDS = {"SGD": 0, "Adagrad": 1, "Adam": 2}

    def __init__(self, optimizer: torch.optim.Optimizer):
        g = optimizer.param_groups[0]
        name = type(optimizer).__name__
        if (name not in self._FLAT_IDS
                or len(optimizer.param_groups) != 1
                or g.get("lr_decay", 0) or g.get("weight_decay", 0)
                or g.get("dampening", 0) or g.get("amsgrad", False)):
            raise ValueError(
                "flat dense path supports plain Adagrad/SGD/Adam (single "
                "group, no weight_decay/lr_decay/dampening/amsgrad)")
        self.kind = name
        self.opt_id = self._FLAT_IDS[name]
        self.lr = g["lr"]
        self.eps = g.get("eps", 1e-10 if name == "Adagrad" else 1e-8)
        self.momentum = float(g.get("momentum", 0.0))
        self.nesterov = bool(g.get("nesterov", False))
        self.betas = g.get("betas", (0.9, 0.999))
        init_acc = g.get("initial_accumulator_value", 0.0)
        params = [p for p in g["params"]
                  if p.requires_grad and p.numel() > 0]
        if not params:
            raise ValueError("no dense params")
        dev = params[0].device
        if any(p.device != dev for p in params):
            raise ValueError("flat dense path needs one device")
        # one flat group per dtype: bf16 params get an fp32 master (the
        # working bf16 weights feed MFMA GEMMs with no per-step casts;
        # accumulator/update math is fp32)
        self.groups = []
        for dt in (torch.float32, torch.bfloat16):
            ps = [p for p in params if p.dtype == dt]
            if not ps:
                continue
            total = sum(p.numel() for p in ps)
            flat = torch.empty(total, device=dev, dtype=dt)
            flat_grad = torch.zeros(total, device=dev, dtype=dt)
            off = 0
            for p in ps:
                n = p.numel()
                flat[off:off + n] = p.data.reshape(-1)
                p.data = flat[off:off + n].view_as(p)
                p.grad = flat_grad[off:off + n].view_as(p)
                off += n
            grp = {
                "dtype": dt,
                "flat": flat,
                "flat_grad": flat_grad,
                "master": (flat.to(torch.float32)
                           if dt == torch.bfloat16 else None),
                "s1": None,
                "s2": None,
            }
            if self.kind == "Adagrad":
                grp["s1"] = torch.full((total,), float(init_acc),
                                       device=dev, dtype=torch.float32)
            elif self.kind == "SGD" and self.momentum:
                grp["s1"] = torch.zeros(total, device=dev,
                                        dtype=torch.float32)
            elif self.kind == "Adam":
                grp["s1"] = torch.zeros(total, device=dev,
                                        dtype=torch.float32)
                grp["s2"] = torch.zeros(total, device=dev,
                                        dtype=torch.float32)
            grp["accum"] = grp["s1"]   # legacy alias (round-1 state dicts)
            self.groups.append(grp)
        # device step counter + Adam bias-correction factors: computed ON
        # DEVICE so a hipGraph-captured step keeps advancing t
        self.step_scalars = (torch.zeros(3, device=dev,
                                         dtype=torch.float32)
                             if self.kind == "Adam" else None)
        self._t = 0      # host mirror for the torch fallback
        self._grads_zeroed = True   # flat_grad starts zeroed
        self.params = params
        # p -> its view into the flat grad buffer (re-bound in zero_grad if
        # model.zero_grad(set_to_none=True) detached it; checked in step)
        self._grad_views = {p: p.grad for p in params}
        self._ext = None
        if dev.type == "cuda":
            from ..ops import require_hip
            self._ext = require_hip()

    def zero_grad(self):
        for p, view in self._grad_views.items():
            if p.grad is not view:
                # model.zero_grad(set_to_none=True) (or a manual p.grad=None)
                # detached the param from the flat buffer; re-bind — later
                # backwards must accumulate into the flat buffer the
                # allreduce and step read
                p.grad = view
        if self._grads_zeroed:
            # step() left the flat buffer zeroed (k_flat_opt writes g[i]=0
            # after consuming it), so the first zero_grad of the next
            # iteration is a no-op — one fewer fill per dtype group inside
            # the captured step. The flag is consumed here: a second
            # zero_grad in the same cycle (e.g. discarding an extra
            # backward's grads) does the real fill.
            self._grads_zeroed = False
            return
        for g in self.groups:
            g["flat_grad"].zero_()

    def _check_bound(self):
        for p, view in self._grad_views.items():
            if p.grad is not view:
                raise RuntimeError(
                    "a dense param's .grad was detached from the flat "
                    "buffer after backward (e.g. model.zero_grad("
                    "set_to_none=True) mid-step); its gradient never "
                    "reached the flat allreduce/step. Call "
                    "optimizer.zero_grad() instead.")

    def step(self):
        self._check_bound()
        self._t += 1
        if self._ext is not None:
            if self.step_scalars is not None:
                self._ext.flat_step_scalars(self.step_scalars,
                                            self.betas[0], self.betas[1])
            if self.kind == "SGD":
                cfg = (self.momentum, 1.0 if self.nesterov else 0.0, 0.0)
            elif self.kind == "Adagrad":
                cfg = (self.eps, 0.0, 0.0)
            else:
                cfg = (self.betas[0], self.betas[1], self.eps)
            for g in self.groups:
                self._ext.flat_opt(self.opt_id, g["flat"], g["master"],
                                   g["s1"], g["s2"], g["flat_grad"],
                                   self.step_scalars, self.lr, *cfg)
            self._grads_zeroed = True   # kernel wrote g[i]=0 after consume
            return
        for g in self.groups:
            grad = g["flat_grad"].to(torch.float32)
            w = g["master"] if g["master"] is not None else g["flat"]
            if self.kind == "Adagrad":
                g["s1"].addcmul_(grad, grad)
                std = g["s1"].sqrt().add_(self.eps)
                w.addcdiv_(grad, std, value=-self.lr)
            elif self.kind == "SGD":
                if self.momentum:
                    g["s1"].mul_(self.momentum).add_(grad)
                    d = (grad + self.momentum * g["s1"]
                         if self.nesterov else g["s1"])
                else:
                    d = grad
                w.add_(d, alpha=-self.lr)
            else:  # Adam
                b1, b2 = self.betas
                g["s1"].mul_(b1).add_(grad, alpha=1 - b1)
                g["s2"].mul_(b2).addcmul_(grad, grad, value=1 - b2)
                mh = g["s1"] / (1 - b1 ** self._t)
                vh = g["s2"] / (1 - b2 ** self._t)
                w.addcdiv_(mh, vh.sqrt().add_(self.eps), value=-self.lr)
            if g["master"] is not None:
                g["flat"].copy_(g["master"])
            g["flat_grad"].zero_()
        self._grads_zeroed = True

    def state_dict(self):
        # detached clones: a caller that keeps training after save must not
        # mutate the saved state through aliased tensors
        return {"groups": [{k: (v.detach().clone()
                                if torch.is_tensor(v) else v)
                            for k, v in g.items()
                            if k not in ("flat_grad", "accum")}
                           for g in self.groups],
                "lr": self.lr, "kind": self.kind, "t": self._t,
                "step_scalars": (self.step_scalars.detach().clone()
                                 if self.step_scalars is not None
                                 else None)}

    def load_state_dict(self, sd):
        for g, s in zip(self.groups, sd["groups"]):
            g["flat"].copy_(s["flat"])
            for k in ("s1", "s2", "master"):
                src = s.get(k, s.get("accum") if k == "s1" else None)
                if g.get(k) is not None and src is not None:
                    g[k].copy_(src)
        self._t = int(sd.get("t", 0))
        if self.step_scalars is not None and sd.get("step_scalars") is not None:
            self.step_scalars.copy_(sd["step_scalars"])


class DistributedOptimizer:
    """Wraps a torch optimizer: dense grads are RCCL-allreduced (SUM, like the
    reference's hvd.DistributedOptimizer(op=hvd.Sum) examples), then the base
    optimizer steps, then every PS storage commits its batch
    (reference exb.py:446-488 distributed optimizer subclasses +
    UpdateWeights op).

    ``flatten_dense=True`` re-bases the dense params onto one flat buffer
    (no-op steady-state zero_grad, single-tensor allreduce, fused flat
    optimizer kernels)."""

    def __init__(self, optimizer: torch.optim.Optimizer,
                 sparse_config: Optional[Dict] = None,
                 average_dense: bool = False,
                 flatten_dense: bool = False):
        self.optimizer = optimizer
        self.average_dense = average_dense
        self._sparse_config = sparse_config
        self._configured_vars = set()
        self.ctx = get_context()
        self._flat: Optional[_FlatDenseOptimizer] = None
        if flatten_dense:
            self._flat = _FlatDenseOptimizer(optimizer)

    # behave like the wrapped optimizer
    def __getattr__(self, name):
        return getattr(self.optimizer, name)

    def zero_grad(self, set_to_none: bool = True):
        if self._flat is not None:
            self._flat.zero_grad()
            return
        self.optimizer.zero_grad(set_to_none=set_to_none)

    def _allreduce_dense(self, async_op: bool = False):
        """Returns a list of work handles when async_op (flat path only) —
        the caller overlaps the sparse commit with the wire time (the
        reference overlapped via update_early_return; here the dense
        allreduce rides alongside the local sparse optimizer kernels)."""
        if not comm.dist_ready() or dist.get_world_size() == 1:
            return []
        if self._flat is not None:
            handles = []
            for g in self._flat.groups:
                h = dist.all_reduce(g["flat_grad"], async_op=async_op)
                if async_op:
                    handles.append((h, g))
                elif self.average_dense:
                    g["flat_grad"] /= dist.get_world_size()
            return handles
        bucket: List[torch.Tensor] = []
        for group in self.optimizer.param_groups:
            for p in group["params"]:
                if p.grad is not None and p.numel() > 0:
                    bucket.append(p.grad)
        if not bucket:
            return
        flat = torch.cat([g.reshape(-1) for g in bucket])
        dist.all_reduce(flat)
        if self.average_dense:
            flat /= dist.get_world_size()
        off = 0
        for g in bucket:
            g.copy_(flat[off:off + g.numel()].view_as(g))
            off += g.numel()

    def _ensure_sparse_configured(self):
        cfg = self._sparse_config
        translate_err = None
        if cfg is None:
            try:
                cfg = _server_cfg_from_torch(self.optimizer)
            except ValueError as e:
                cfg = None
                translate_err = e
        for e in _tracked_live():
            if e.variable is None:
                continue
            vid = e.variable.sharded.variable_id
            if vid in self._configured_vars:
                continue
            if e.variable.sharded.shard.optimizer is None:
                if cfg is None:
                    raise RuntimeError(
                        "no sparse optimizer configured; pass sparse_config= "
                        "to distributed_optimizer or call set_optimizer on the "
                        "Embedding variable"
                        + (f" ({translate_err})" if translate_err else ""))
                c = dict(cfg)
                cat = c.pop("category")
                e.variable.set_optimizer(cat, **c)
            self._configured_vars.add(vid)

    def step(self, closure=None):
        loss = None
        if closure is not None:
            loss = closure()
        if self._flat is not None:
            # overlap: dense allreduce in flight while the sparse commit's
            # local optimizer kernels run; wait, then dense step
            handles = self._allreduce_dense(async_op=True)
            self._ensure_sparse_configured()
            self.ctx.update_all_weights()
            for h, g in handles:
                h.wait()
                if self.average_dense:
                    g["flat_grad"] /= dist.get_world_size()
            self._flat.step()
        else:
            self._allreduce_dense()
            self.optimizer.step()
            self._ensure_sparse_configured()
            self.ctx.update_all_weights()
        self.ctx.model_version += 1
        return loss

    def state_dict(self):
        if self._flat is not None:
            return {"flat_dense": self._flat.state_dict()}
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        if self._flat is not None and "flat_dense" in sd:
            self._flat.load_state_dict(sd["flat_dense"])
            return
        self.optimizer.load_state_dict(sd)


def distributed_optimizer(optimizer: torch.optim.Optimizer,
                          sparse_config: Optional[Dict] = None,
                          **kw) -> DistributedOptimizer:
    return DistributedOptimizer(optimizer, sparse_config=sparse_config, **kw)


def _make_opt_class(torch_name: str):
    """Pre-wrapped optimizer constructors (the reference exported
    distributed subclasses of all keras optimizers, exb.py:446-488):
    ``embed.Adagrad(model.parameters(), lr=0.01)`` ==
    ``distributed_optimizer(torch.optim.Adagrad(...))``."""
    base = getattr(torch.optim, torch_name)

    def ctor(params, *args, sparse_config: Optional[Dict] = None,
             flatten_dense: bool = False, **kw):
        return distributed_optimizer(base(params, *args, **kw),
                                     sparse_config=sparse_config,
                                     flatten_dense=flatten_dense)

    ctor.__name__ = torch_name
    ctor.__qualname__ = torch_name
    return ctor


Adadelta = _make_opt_class("Adadelta")
Adagrad = _make_opt_class("Adagrad")
Adam = _make_opt_class("Adam")
Adamax = _make_opt_class("Adamax")
NAdam = Nadam = _make_opt_class("NAdam")
RMSprop = _make_opt_class("RMSprop")
SGD = _make_opt_class("SGD")


def Ftrl(params, sparse_config: Optional[Dict] = None, **kw):
    """FTRL exists only as a SERVER-side sparse optimizer (torch has no
    dense FTRL; the reference's keras Ftrl also only mattered server-side).
    Dense params train with Adagrad; the sparse side runs the engine's FTRL
    (core/optimizers.py, exact reference formulas incl. l2_shrinkage and
    lr_power branches)."""
    cfg = dict(category="ftrl",
               learning_rate=kw.pop("learning_rate", kw.pop("lr", 0.05)),
               l1_regularization_strength=kw.pop(
                   "l1_regularization_strength", 0.0),
               l2_regularization_strength=kw.pop(
                   "l2_regularization_strength", 0.0))
    if sparse_config:
        cfg.update(sparse_config)
    return distributed_optimizer(
        torch.optim.Adagrad(params, lr=cfg["learning_rate"], **kw),
        sparse_config=cfg)


# ----------------------------------------------------------- model helpers

def distributed_model(model: nn.Module, sparse_as_dense_size: int = 64
                      ) -> nn.Module:
    """Replace every nn.Embedding in ``model`` with a PS-backed Embedding
    (vocab < sparse_as_dense_size stays a replicated dense table — the
    reference's policy, exb.py:241-248,593-642). In-place; returns model."""
    for name, mod in list(model.named_children()):
        if isinstance(mod, nn.Embedding):
            if mod.num_embeddings < sparse_as_dense_size:
                continue
            rep = Embedding(mod.num_embeddings, mod.embedding_dim)
            setattr(model, name, rep)
        else:
            distributed_model(mod, sparse_as_dense_size)
    return model


def sparse_read_as_dense(variable: Variable, vocab: int,
                         readonly: bool = False) -> torch.Tensor:
    """Materialize rows [0, vocab) of a variable on every rank
    (reference sparse_read_as_dense / save_as_original_model bulk pull).

    Default readonly=False matches the reference: its export pulled
    through the TRAINING path, so never-touched rows materialize with
    their initializer values (exb.py:529-538), not zeros."""
    chunks = []
    step = max(1, (1 << 20) // max(1, variable.embedding_dim))
    for start in range(0, vocab, step):
        n = min(step, vocab - start)
        chunks.append(variable.sharded.pull_dense(start, n,
                                                  readonly=readonly))
    return torch.cat(chunks)


# ------------------------------------------------------- checkpoint surface

def save_server_model(path: str, include_optimizer: bool = True):
    from ..checkpoint import dump_model
    dump_model(get_context(), path, include_optimizer=include_optimizer)


def load_server_model(path: str):
    from ..checkpoint import load_model
    load_model(get_context(), path)


def save_as_original_model(model: nn.Module, path: str,
                           format: str = "saved_model"):
    """Export a fully materialized, PS-free copy of the model (reference
    exb.py:506-547 clone-and-materialize).

    format="saved_model" (default, the north-star contract): a TensorFlow
    SavedModel directory (saved_model.pb + variables bundle) that TF /
    TF-Serving loads with no OpenEmbedding runtime — supported for the
    model zoo (LR/WDL/DeepFM/xDeepFM); see export_tf.py.

    format="torch": a torch.save state dict with every PS Embedding
    materialized into a dense weight; loadable without openembedding_amd.
    """
    if format == "saved_model":
        from ..export_tf import export_saved_model
        try:
            return export_saved_model(model, path)
        except (NotImplementedError, AttributeError) as e:
            raise NotImplementedError(
                f"SavedModel export covers the model zoo "
                f"(LR/WDL/DeepFM/xDeepFM); for custom modules use "
                f"save_as_original_model(..., format='torch') ({e})")
    if format != "torch":
        raise ValueError(f"unknown export format {format!r}")
    state = {}
    for name, mod in model.named_modules():
        if isinstance(mod, Embedding) and not mod.sparse_as_dense:
            if mod.num_embeddings < 0:
                raise ValueError("cannot materialize a hash-mode Embedding "
                                 "into a dense table")
            rows = sparse_read_as_dense(mod.variable, mod.num_embeddings)
            state[name + ".weight"] = rows.cpu()
    sd = {k: v.cpu() for k, v in model.state_dict().items()
          if not k.endswith("grad_hook")}
    sd.update(state)
    if get_context().rank == 0:
        torch.save({"state_dict": sd, "format": "openembedding_amd.original",
                    "version": "0.2"}, path)


class Model(nn.Module):
    """Wrapper adding server-model save/load beside the torch checkpoint
    (reference _DistributedModel, exb.py:550-583)."""

    def __init__(self, module: nn.Module):
        super().__init__()
        self.module = module

    def forward(self, *a, **kw):
        return self.module(*a, **kw)

    def save_weights(self, filepath: str):
        if get_context().rank == 0:
            torch.save(self.module.state_dict(), filepath)
        save_server_model(filepath + ".openembedding")

    def load_weights(self, filepath: str):
        sd = torch.load(filepath, map_location=get_context().device,
                        weights_only=True)
        self.module.load_state_dict(sd, strict=False)
        load_server_model(filepath + ".openembedding")


# ------------------------------------------------------------ prefetch API

def pulling(loader, model=None, depth: int = 2, sparse_index: int = 1,
            getter=None):
    """Dataset-side prefetch pipeline (reference pulling(), exb.py:645-691 +
    PrefetchPullWeights): while batch t trains, the embedding pulls of
    batches t+1..t+depth are already issued on the prefetch stream, so the
    unique/all_to_all/gather work overlaps the MLP compute.

    Batch -> indices mapping: ``getter(batch) -> {embedding_module: indices}``
    or by default every tracked PS Embedding of ``model`` gets
    ``batch[sparse_index]`` (the (dense, sparse, labels) convention of the
    model zoo; the reference matched model input layers by name,
    exb.py:656-682).

    The prefetched result is matched by tensor identity at forward time, so
    the SAME batch objects yielded here must flow into the model."""
    import collections

    if getter is None:
        embs = ([m for m in model.modules()
                 if isinstance(m, Embedding) and not m.sparse_as_dense]
                if model is not None else [])

        def getter(batch):
            return {e: batch[sparse_index] for e in embs}

    it = iter(loader)
    buf = collections.deque()

    def fill():
        while len(buf) < depth + 1:
            try:
                buf.append(next(it))
            except StopIteration:
                return

    def issue(b):
        for emb, idx in getter(b).items():
            emb.prefetch(idx)

    # Ordering guarantee (the reference built it from the server-side
    # batch-id pending queue, EmbeddingPullOperator.cpp:125-141): a batch's
    # pull must see the PREVIOUS batch's commit. Issuing the prefetch for
    # batch t+1 only after batch t was trained puts it after the commit in
    # stream order — exactly one batch of pipelining, the steady state the
    # reference's deferral also converges to. ``depth`` only sizes the
    # host-side loader read-ahead.
    fill()
    if buf:
        issue(buf[0])  # first batch: nothing committed yet
    while buf:
        yield buf.popleft()
        fill()
        if buf:
            issue(buf[0])  # after the just-trained batch's commit


# ------------------------------------------------------------- PMem parity

def should_persist_server_model() -> bool:
    """The cache-full backpressure signal of the capacity tier (reference
    should_persist, PmemEmbeddingOptimizerVariable.h:84-86 propagated
    through EmbeddingPullOperator.cpp:182-189): True when any tiered
    variable's device cache filled since the last persist."""
    ctx = get_context()
    return any(getattr(v.shard, "should_persist", lambda: False)()
               for v in ctx.variables.values())


def persist_server_model(path: str):
    """Lightweight checkpoint of the capacity tier (reference
    exb_persist_model): flush dirty cache rows to the host tier, dump, and
    commit the watermark. For untiered variables this equals
    save_server_model."""
    ctx = get_context()
    for v in ctx.variables.values():
        if hasattr(v.shard, "persist"):
            v.shard.persist()
    save_server_model(path, include_optimizer=True)
    for v in ctx.variables.values():
        if hasattr(v.shard, "checkpoint_committed"):
            v.shard.checkpoint_committed()


def restore_server_model(path: str):
    load_server_model(path)
