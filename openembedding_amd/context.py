"""Worker context: process-group bootstrap + variable/storage registry.

Replaces the reference's whole coordination substrate — master daemon, RPC
Communication barrier/broadcast, WorkerContext, Connection
(reference client/WorkerContext.cpp, client/Connection.cpp, Communication.cpp)
— with torch.distributed: rendezvous comes from torchrun env vars, control
collectives are dist.barrier/broadcast, and the "server" is the local shard
engine on this rank's GPU (embedded-server default of the reference,
openembedding/__init__.py:57-76).

Storage/variable creation is COLLECTIVE: every rank must call create_storage /
create_variable in the same order (the reference enforced the same with
sync_bcast rendezvous, WorkerContext.cpp:66-113; here same-order calls give the
same ids deterministically, no RPC needed).
"""

from __future__ import annotations

import atexit
import os
import time
import uuid
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from . import flags
from .config import EnvConfig
from .core.variable import VariableMeta, VariableShard, HASH_VOCAB_THRESHOLD
from .parallel.sharded import ShardedVariable
from .utils.metrics import Reporter

_context: Optional["Context"] = None


def _dist_ready() -> bool:
    return dist.is_available() and dist.is_initialized()


class Storage:
    """A group of variables committed together (reference EmbeddingStorage.h).

    ``update_weights`` commits every variable's pending gradients — the
    reference's UDF "store" operator (EmbeddingStoreOperator.cpp:23-81); the
    pending-batch ordering machinery is unnecessary here because collectives
    and kernels on one stream are already ordered."""

    def __init__(self, ctx: "Context", storage_id: int, num_shards: int):
        self.ctx = ctx
        self.storage_id = storage_id
        self.num_shards = num_shards
        self.variables: List[ShardedVariable] = []
        self.batch_id = 0

    def create_variable(self, vocabulary_size: int, embedding_dim: int,
                        dtype: torch.dtype = torch.float32) -> ShardedVariable:
        var = self.ctx._create_variable(self, vocabulary_size, embedding_dim, dtype)
        self.variables.append(var)
        return var

    def update_weights(self) -> None:
        for v in self.variables:
            v.update_weights()
        self.batch_id += 1


class Context:
    """Per-process session (reference WorkerContext + exb.py Context)."""

    def __init__(self, device: Optional[str] = None, seed: int = 0):
        self.rank = int(os.environ.get("RANK", 0))
        self.world_size = int(os.environ.get("WORLD_SIZE", 1))
        self.local_rank = int(os.environ.get("LOCAL_RANK", self.rank))
        if self.world_size > 1 and not _dist_ready():
            # OEAMD_BACKEND=gloo on a GPU box runs the full engine with the
            # CPU-staged wire (debug / N-ranks-on-1-GPU rehearsal — RCCL
            # refuses duplicate devices in one communicator)
            backend = (os.environ.get("OEAMD_BACKEND")
                       or ("nccl" if torch.cuda.is_available() else "gloo"))
            if backend == "nccl":
                torch.cuda.set_device(self.local_rank)
            dist.init_process_group(backend=backend)
            self._owns_pg = True
        else:
            self._owns_pg = False
        if _dist_ready():
            self.rank = dist.get_rank()
            self.world_size = dist.get_world_size()
        if device is None:
            device = os.environ.get("OEAMD_DEVICE") or (
                f"cuda:{self.local_rank}" if torch.cuda.is_available()
                else "cpu")
        if device.startswith("cuda"):
            torch.cuda.set_device(device)
        self.device = torch.device(device)
        self.seed = seed
        self.storages: List[Storage] = []
        self.variables: Dict[int, ShardedVariable] = {}
        self._next_variable_id = 0
        # model uuid: rank-0 generated, broadcast (reference py_api.cc:92-98)
        if _dist_ready():
            obj = [uuid.uuid4().hex if self.rank == 0 else None]
            dist.broadcast_object_list(obj, src=0)
            self.model_uuid = obj[0]
        else:
            self.model_uuid = uuid.uuid4().hex
        self.model_version = 0
        self._t0 = time.time()
        # typed config tree from flags.config (reference EnvConfig via
        # embed.flags.config YAML, openembedding/__init__.py:8-41)
        self.config = EnvConfig.parse(flags.config)
        self._reporter = Reporter(self.config.server.report_interval,
                                  rank=self.rank)
        self._reporter.start()
        self._prefetch_stream = None

    @property
    def prefetch_stream(self):
        """Side HIP stream for ahead-of-time pulls (reference dataset-thread
        prefetch; here overlap is stream-level)."""
        if self._prefetch_stream is None:
            self._prefetch_stream = torch.cuda.Stream()
        return self._prefetch_stream

    # ------------------------------------------------------------- factories

    def create_storage(self, num_shards: int = -1) -> Storage:
        if num_shards is None or num_shards <= 0:
            num_shards = self.world_size
        st = Storage(self, len(self.storages), num_shards)
        self.storages.append(st)
        return st

    def _create_variable(self, storage: Storage, vocabulary_size: int,
                         embedding_dim: int, dtype: torch.dtype) -> ShardedVariable:
        vid = self._next_variable_id
        self._next_variable_id += 1
        if vocabulary_size is None or vocabulary_size < 0:
            vocabulary_size = HASH_VOCAB_THRESHOLD
        meta = VariableMeta(variable_id=vid, embedding_dim=embedding_dim,
                            dtype=dtype, vocabulary_size=vocabulary_size)
        cache_mb = self.config.server.cache_size_mb
        if self.device.type == "cuda" and dtype == torch.float64:
            # f64 parity (reference registers f32 AND f64 variables,
            # EmbeddingVariable.cpp:277-279): the HIP kernels are f32-only,
            # so f64 tables run the torch-op engine on the same cuda device
            # — correct, slower, and documented as such
            shard_cls = VariableShard
            kw = {}
        elif self.device.type == "cuda":
            if cache_mb > 0 and meta.use_hash_table:
                from .core.tiered_gpu import HipTieredVariableShard
                row_bytes = 4 * (embedding_dim + 64)
                shard_cls = HipTieredVariableShard
                kw = {"cache_rows": max(1024, (cache_mb << 20) // row_bytes)}
            else:
                from .core.variable_gpu import HipVariableShard
                shard_cls = HipVariableShard
                kw = {}
        elif cache_mb > 0 and meta.use_hash_table:
            from .core.tiered import TieredVariableShard
            row_bytes = 4 * (embedding_dim + 64)  # dim + worst-case state
            shard_cls = TieredVariableShard
            kw = {"cache_rows": max(1024, (cache_mb << 20) // row_bytes)}
        else:
            shard_cls = VariableShard
            kw = {}
        shard = shard_cls(meta, shard_id=self.rank,
                          shard_num=self.world_size,
                          device=str(self.device), seed=self.seed, **kw)
        var = ShardedVariable(shard, storage)
        self.variables[vid] = var
        return var

    # ------------------------------------------------------------- collectives

    def barrier(self) -> None:
        if _dist_ready():
            dist.barrier()

    def broadcast_object(self, obj, src: int = 0):
        if not _dist_ready():
            return obj
        box = [obj]
        dist.broadcast_object_list(box, src=src)
        return box[0]

    def update_all_weights(self) -> None:
        for st in self.storages:
            st.update_weights()

    def finalize(self) -> None:
        global _context
        self._reporter.stop()
        if self._owns_pg and _dist_ready():
            dist.destroy_process_group()
        if _context is self:
            _context = None


def get_context(device: Optional[str] = None) -> Context:
    """The process-wide context singleton (reference exb.py:107-148)."""
    global _context
    if _context is None:
        _context = Context(device=device)
        atexit.register(_shutdown)
    return _context


def _shutdown():
    global _context
    if _context is not None:
        try:
            _context.finalize()
        except Exception:
            pass
        _context = None
