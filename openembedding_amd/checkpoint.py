"""Checkpoint dump/load — the reference's portable model format, rebuilt.

Logical layout parity with the reference (SURVEY.md §3.4, §5 Checkpoint):
  <uri>/model_meta                        JSON: sign, variables, version "0.2"
        (reference Model.cpp:89-108, Meta.h:105-145)
  <uri>/<storage_ordinal>/model_<rank>_0  binary shard file: per variable a
        JSON header + blocks of {n, indices[n], weights[n*dim],
        states[n*state_dim]} (reference EmbeddingDumpOperator.cpp:57-96,
        EmbeddingShardFile.h:13-26)

Differences from the reference, by design:
  - keys are stored GLOBAL (the reference stores shard-local indices and
    re-globalizes on load via idx*shard_num+shard_id,
    EmbeddingShardFile.h:23-25); global keys make the file shard-count
    independent by construction — load re-shards by key % world_size exactly
    like the reference's init-push path (EmbeddingLoadOperator.cpp:58-111);
  - serialization is plain little-endian numpy buffers with a JSON header
    (the reference's BinaryArchive lives in the absent pico-core submodule,
    so its byte format is not reproducible, only its structure).

Every rank writes its own shard file (collective dump); load is collective,
each rank reads all files and keeps its own keys.
"""

from __future__ import annotations

import json
import os
import struct
from typing import Optional

import numpy as np
import torch

MAGIC = b"OEAMDSH1"
BLOCK_ROWS = 1 << 16  # rows per block (~ the reference's ~1MiB batching,
                      # EmbeddingVariable.cpp:84-87)


def _var_header(var, include_optimizer: bool) -> dict:
    shard = var.shard
    return {
        "variable_id": shard.meta.variable_id,
        "datatype": shard.meta.datatype_str(),
        "embedding_dim": shard.dim,
        "vocabulary_size": shard.meta.vocabulary_size,
        "state_dim": shard.state_dim if include_optimizer else 0,
        "optimizer": ({"category": shard.optimizer.category,
                       **shard.optimizer.dump_config()}
                      if (include_optimizer and shard.optimizer) else None),
        "initializer": {"category": shard.initializer.category,
                        **shard.initializer.dump_config()},
        "shard_id": shard.shard_id,
        "shard_num": shard.shard_num,
    }


def dump_model(ctx, uri: str, include_optimizer: bool = True) -> None:
    """Collective: every rank writes its shard of every storage."""
    rank, world = ctx.rank, ctx.world_size
    meta = {
        "model_sign": f"{ctx.model_uuid}-{ctx.model_version}",
        "version": "0.2",
        "num_storages": len(ctx.storages),
        "world_size": world,
        "variables": [],
    }
    for st in ctx.storages:
        for var in st.variables:
            meta["variables"].append({
                "variable_id": var.shard.meta.variable_id,
                "storage_ordinal": st.storage_id,
                # reference ModelVariableMeta field (Meta.h:77-99): with it
                # present, this JSON is a superset of the reference's
                # ModelOfflineMeta schema (model_sign + variables[datatype,
                # embedding_dim, vocabulary_size, storage_name] + version)
                "storage_name": str(st.storage_id),
                "datatype": var.shard.meta.datatype_str(),
                "embedding_dim": var.shard.dim,
                "vocabulary_size": var.shard.meta.vocabulary_size,
            })
    if rank == 0:
        os.makedirs(uri, exist_ok=True)
        for st in ctx.storages:
            os.makedirs(os.path.join(uri, str(st.storage_id)), exist_ok=True)
        with open(os.path.join(uri, "model_meta"), "w") as f:
            json.dump(meta, f, indent=1)
    ctx.barrier()
    # files per rank: the reference's server.server_dump_files knob
    # (EnvConfig.cpp; files named model_{node}_{file_id},
    # EmbeddingDumpOperator.cpp:28). Rows round-robin across files so big
    # shards split for parallel HDFS-style upload.
    n_files = max(1, getattr(ctx.config.server, "server_dump_files", 1)
                  if getattr(ctx, "config", None) else 1)
    for st in ctx.storages:
        files = [open(os.path.join(uri, str(st.storage_id),
                                   f"model_{rank}_{i}"), "wb")
                 for i in range(n_files)]
        try:
            for j, var in enumerate(st.variables):
                _dump_variable(files[j % n_files], var, include_optimizer)
        finally:
            for f in files:
                f.close()
    ctx.barrier()


def _dump_variable(f, var, include_optimizer: bool) -> None:
    keys, w, s = var.shard.export_rows(include_state=include_optimizer)
    hdr = _var_header(var, include_optimizer and s is not None)
    hdr["num_items"] = int(keys.numel())
    hjson = json.dumps(hdr).encode()
    f.write(MAGIC)
    f.write(struct.pack("<q", len(hjson)))
    f.write(hjson)
    n = keys.numel()
    sd = hdr["state_dim"]
    for start in range(0, max(n, 1), BLOCK_ROWS):
        if n == 0:
            f.write(struct.pack("<q", 0))
            break
        stop = min(start + BLOCK_ROWS, n)
        blk = stop - start
        f.write(struct.pack("<q", blk))
        f.write(keys[start:stop].cpu().numpy().astype("<i8").tobytes())
        f.write(w[start:stop].cpu().numpy().astype("<f4").tobytes())
        if sd:
            f.write(s[start:stop].cpu().numpy().astype("<f4").tobytes())
    f.write(struct.pack("<q", -1))  # end of variable section


def _apply_config(var, hdr: dict) -> None:
    """Restore the variable's initializer/optimizer config from the shard
    header (the reference ships init_config with variable creation and
    re-applies it on load, EmbeddingInitOperator.cpp:138-168)."""
    init = hdr.get("initializer")
    if init:
        c = dict(init)
        cat = c.pop("category")
        var.shard.set_initializer(cat, **c)
    opt = hdr.get("optimizer")
    if opt:
        c = dict(opt)
        cat = c.pop("category")
        cur = var.shard.optimizer
        if cur is None or cur.category != cat or cur.dump_config() != c:
            var.shard.set_optimizer(cat, **c)


def read_meta(uri: str) -> dict:
    with open(os.path.join(uri, "model_meta")) as f:
        return json.load(f)


def iter_blocks(uri: str, meta: Optional[dict] = None):
    """Stream every shard file of a dump: yields (header_dict, keys <i8 [n],
    weights <f4 [n,dim], states <f4 [n,sd] or None) per block. The streaming
    analogue of the reference's EmbeddingLoadOperator generate_push_items
    (EmbeddingLoadOperator.cpp:58-111)."""
    meta = meta or read_meta(uri)
    for st_ord in range(meta["num_storages"]):
        d = os.path.join(uri, str(st_ord))
        for fname in sorted(os.listdir(d)):
            if not fname.startswith("model_"):
                continue
            with open(os.path.join(d, fname), "rb") as f:
                while True:
                    magic = f.read(8)
                    if not magic:
                        break
                    if magic != MAGIC:
                        raise RuntimeError(f"bad shard file {fname}")
                    (hlen,) = struct.unpack("<q", f.read(8))
                    hdr = json.loads(f.read(hlen))
                    dim = hdr["embedding_dim"]
                    sd = hdr["state_dim"]
                    while True:
                        (blk,) = struct.unpack("<q", f.read(8))
                        if blk <= 0:
                            if blk == 0:
                                (end,) = struct.unpack("<q", f.read(8))
                                assert end == -1
                            break
                        keys = np.frombuffer(f.read(blk * 8), dtype="<i8")
                        w = np.frombuffer(f.read(blk * dim * 4),
                                          dtype="<f4").reshape(blk, dim)
                        s = None
                        if sd:
                            s = np.frombuffer(f.read(blk * sd * 4),
                                              dtype="<f4").reshape(blk, sd)
                        yield hdr, keys, w, s


def load_model(ctx, uri: str) -> None:
    """Collective: clears all variables, streams every shard file and keeps
    the keys this rank owns (key % world == rank), re-sharding like the
    reference's load-through-init-push (EmbeddingLoadOperator.cpp:58-111)."""
    meta = read_meta(uri)
    by_id = {}
    for st in ctx.storages:
        for var in st.variables:
            by_id[var.shard.meta.variable_id] = var
    for mvar in meta["variables"]:
        vid = mvar["variable_id"]
        if vid not in by_id:
            raise RuntimeError(f"checkpoint has variable {vid} missing in model")
        v = by_id[vid]
        if v.shard.dim != mvar["embedding_dim"]:
            raise RuntimeError(f"variable {vid}: dim mismatch "
                               f"{v.shard.dim} != {mvar['embedding_dim']}")
        v.shard.clear()
    rank, world = ctx.rank, ctx.world_size
    configured = set()
    for hdr, keys, w, s in iter_blocks(uri, meta):
        var = by_id.get(hdr["variable_id"])
        if var is None:
            continue
        if hdr["variable_id"] not in configured:
            configured.add(hdr["variable_id"])
            _apply_config(var, hdr)
        sd = hdr["state_dim"]
        mine = (keys % world) == rank
        if not mine.any():
            continue
        kt = torch.from_numpy(keys[mine].copy()).to(ctx.device)
        wt = torch.from_numpy(w[mine].copy()).to(ctx.device)
        st_t: Optional[torch.Tensor] = None
        if s is not None and sd == var.shard.state_dim:
            st_t = torch.from_numpy(s[mine].copy()).to(ctx.device)
        var.shard.import_rows(kt, wt, st_t)
    ctx.barrier()
