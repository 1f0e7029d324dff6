"""Serving: model registry, controller and REST API.

MI355X rebuild of the reference's serving stack:
- ``ModelManager::find_model_variable`` (reference
  client/ModelController.cpp:24-44): sign -> loaded model -> read-only
  variable handle, cached -> :class:`ModelManager.find_model_variable`;
- ``ModelController`` create/delete/show models, show/shutdown nodes with a
  CREATING -> NORMAL status machine and async heavy load (reference
  ModelController.cpp:47-164) -> :class:`ModelController`;
- brpc REST controller ``POST/GET/DELETE /models[/sign]``,
  ``GET/DELETE /nodes[/id]`` (reference entry/controller.cc:100-203) ->
  :func:`make_app` (FastAPI; run with uvicorn).

Architectural difference, by design: the reference serves from live PS
server processes holding training shards with replica HA; here a serving
process loads the dump into its own read-only tables (HBM on a GPU box, host
memory otherwise) — single-node MI355X serving needs no RPC fabric, and HA
across boxes is a deployment concern (run N serving processes behind a load
balancer). Missing keys pull zeros, exactly like the reference read-only
path (EmbeddingPullOperator.cpp:179-181 get_weights).
"""

# NOTE: no `from __future__ import annotations` here — FastAPI must see real
# class objects (not strings) for the locally-defined request models in
# make_app, or it demotes the body params to query params.
import threading
import time
from typing import Dict, List, Optional

import numpy as np
import torch

from .checkpoint import iter_blocks, read_meta
from .core.variable import VariableMeta, VariableShard


class ModelStatus:
    CREATING = "CREATING"
    NORMAL = "NORMAL"
    DELETED = "DELETED"
    ERROR = "ERROR"


class ServedVariable:
    """Read-only handle over one loaded variable (reference
    EmbeddingVariableHandle in read-only mode)."""

    def __init__(self, shard: VariableShard):
        self.shard = shard

    @property
    def embedding_dim(self) -> int:
        return self.shard.dim

    def pull_weights(self, indices: torch.Tensor) -> torch.Tensor:
        flat = indices.reshape(-1).to(torch.int64).to(self.shard.device)
        out = self.shard.pull_readonly(flat)
        return out.view(*indices.shape, self.shard.dim)


class ServedModel:
    """One loaded model: meta + read-only variables keyed by variable_id."""

    def __init__(self, sign: str, uri: str):
        self.sign = sign
        self.uri = uri
        self.status = ModelStatus.CREATING
        self.error: Optional[str] = None
        self.created_at = time.time()
        self.variables: Dict[int, ServedVariable] = {}
        self.meta: Optional[dict] = None

    def load(self, device: str = "cpu") -> None:
        """Stream the dump into read-only tables. All shards of the dump are
        merged into one local table (shard_num=1 view — serving is per-key
        lookup, the training shard layout is irrelevant).

        On a cuda device the tables are HipVariableShards, so serving pulls
        run the same k_ht_lookup / k_gather_init HIP kernels as training
        (the round-1 CPU-table fallback bypassed every kernel)."""
        meta = read_meta(self.uri)
        self.meta = meta
        shard_cls = VariableShard
        if device.startswith("cuda"):
            from .core.variable_gpu import HipVariableShard
            shard_cls = HipVariableShard
        shards: Dict[int, VariableShard] = {}
        for mvar in meta["variables"]:
            vm = VariableMeta(variable_id=mvar["variable_id"],
                              embedding_dim=mvar["embedding_dim"],
                              vocabulary_size=mvar["vocabulary_size"])
            shards[vm.variable_id] = shard_cls(vm, shard_id=0, shard_num=1,
                                               device=device)
        for hdr, keys, w, _s in iter_blocks(self.uri, meta):
            sh = shards.get(hdr["variable_id"])
            if sh is None or not len(keys):
                continue
            # frombuffer arrays are read-only; copy before wrapping (torch
            # tensors over read-only memory are undefined behavior)
            kt = torch.from_numpy(keys.copy())
            wt = torch.from_numpy(w.copy())
            sh.import_rows(kt.to(sh.device), wt.to(sh.device))
        self.variables = {vid: ServedVariable(sh)
                          for vid, sh in shards.items()}
        self.status = ModelStatus.NORMAL

    def export_block(self, variable_id: int, start: int, count: int):
        """Row block [start, start+count) of a variable in export order —
        the replica-to-replica restore wire (reference
        EmbeddingRestoreOperator.cpp:19-106 pulled shard content from live
        replicas in server_block_num_items batches; here the batches ride
        HTTP). The export snapshot is cached on first use (tables are
        read-only while serving)."""
        if not hasattr(self, "_export_cache"):
            self._export_cache = {}
        blk = self._export_cache.get(variable_id)
        if blk is None:
            sh = self.variables[variable_id].shard
            keys, w, _s = sh.export_rows(include_state=False)
            blk = (keys.cpu(), w.cpu())
            self._export_cache[variable_id] = blk
        keys, w = blk
        end = min(start + count, keys.numel())
        return {"total": int(keys.numel()),
                "keys": keys[start:end].tolist(),
                "weights": w[start:end].tolist()}

    def load_from_replica(self, base_url: str, sign: str,
                          device: str = "cpu",
                          block: int = 65536) -> None:
        """Coordinated restore: rebuild this model's tables by paging rows
        out of a LIVE replica (no dump needed — the reference's
        restore-from-replica path for a replacement server)."""
        import requests

        meta = requests.get(f"{base_url}/models/{sign}", timeout=10).json()
        self.meta = {"model_sign": meta["model_sign"],
                     "variables": meta["variables"], "version": "0.2"}
        shard_cls = VariableShard
        if device.startswith("cuda"):
            from .core.variable_gpu import HipVariableShard
            shard_cls = HipVariableShard
        shards: Dict[int, VariableShard] = {}
        for mvar in meta["variables"]:
            vm = VariableMeta(variable_id=mvar["variable_id"],
                              embedding_dim=mvar["embedding_dim"],
                              vocabulary_size=mvar["vocabulary_size"])
            sh = shard_cls(vm, shard_id=0, shard_num=1, device=device)
            start = 0
            while True:
                r = requests.get(
                    f"{base_url}/models/{sign}/variables/"
                    f"{vm.variable_id}/rows",
                    params={"start": start, "count": block},
                    timeout=60).json()
                if r["keys"]:
                    kt = torch.tensor(r["keys"], dtype=torch.int64,
                                      device=sh.device)
                    wt = torch.tensor(r["weights"], dtype=torch.float32,
                                      device=sh.device)
                    sh.import_rows(kt, wt)
                start += block
                if start >= r["total"]:
                    break
            shards[vm.variable_id] = sh
        self.variables = {vid: ServedVariable(sh)
                          for vid, sh in shards.items()}
        self.status = ModelStatus.NORMAL

    def describe(self) -> dict:
        return {
            "model_sign": self.sign,
            "uri": self.uri,
            "status": self.status,
            "error": self.error,
            "variables": ([
                {"variable_id": v["variable_id"],
                 "embedding_dim": v["embedding_dim"],
                 "vocabulary_size": v["vocabulary_size"]}
                for v in self.meta["variables"]] if self.meta else []),
        }


class ModelManager:
    """sign -> ServedModel cache + variable resolution (reference
    ModelManager::find_model_variable, ModelController.cpp:24-44)."""

    def __init__(self, device: str = "cpu"):
        self.device = device
        self._models: Dict[str, ServedModel] = {}
        self._lock = threading.Lock()

    def get(self, sign: str) -> Optional[ServedModel]:
        return self._models.get(sign)

    def find_model_variable(self, sign: str, variable_id: int
                            ) -> ServedVariable:
        m = self._models.get(sign)
        if m is None:
            raise KeyError(f"model {sign!r} not found")
        if m.status != ModelStatus.NORMAL:
            raise RuntimeError(f"model {sign!r} status {m.status}")
        v = m.variables.get(variable_id)
        if v is None:
            raise KeyError(f"model {sign!r} has no variable {variable_id}")
        return v

    def _register(self, model: ServedModel) -> None:
        with self._lock:
            self._models[model.sign] = model

    def _remove(self, sign: str) -> Optional[ServedModel]:
        with self._lock:
            return self._models.pop(sign, None)

    def signs(self) -> List[str]:
        return sorted(self._models)


class ModelController:
    """Create/delete/show models; node surface (reference
    ModelController.cpp:47-164 with the master-lock + async-load machinery
    collapsed to a thread per load — one process owns the registry)."""

    def __init__(self, manager: Optional[ModelManager] = None,
                 device: str = "cpu"):
        self.manager = manager or ModelManager(device=device)
        self.shutdown_requested = False

    def create_model(self, model_uri: str, sign: Optional[str] = None,
                     wait: bool = True) -> ServedModel:
        """Load a dump for serving. sign defaults to the dump's model_sign.
        wait=False returns immediately with status CREATING (reference async
        heavy load, ModelController.cpp:66-82)."""
        meta = read_meta(model_uri)
        sign = sign or meta["model_sign"]
        existing = self.manager.get(sign)
        if existing is not None and existing.status != ModelStatus.ERROR:
            raise ValueError(f"model {sign!r} already exists")
        model = ServedModel(sign, model_uri)
        self.manager._register(model)

        def _load():
            try:
                model.load(device=self.manager.device)
            except Exception as e:  # noqa: BLE001
                model.status = ModelStatus.ERROR
                model.error = repr(e)

        if wait:
            _load()
            if model.status == ModelStatus.ERROR:
                self.manager._remove(sign)
                raise RuntimeError(f"load failed: {model.error}")
        else:
            threading.Thread(target=_load, daemon=True).start()
        return model

    def restore_model_from_replica(self, base_url: str, sign: str,
                                   wait: bool = True) -> ServedModel:
        """Coordinated restore from a live replica (reference
        EmbeddingRestoreOperator replica path / server --restore): page
        the rows out of ``base_url`` instead of a dump URI."""
        existing = self.manager.get(sign)
        if existing is not None and existing.status != ModelStatus.ERROR:
            raise ValueError(f"model {sign!r} already exists")
        model = ServedModel(sign, f"replica://{base_url}")
        self.manager._register(model)

        def _load():
            try:
                model.load_from_replica(base_url, sign,
                                        device=self.manager.device)
            except Exception as e:  # noqa: BLE001
                model.status = ModelStatus.ERROR
                model.error = repr(e)

        if wait:
            _load()
            if model.status == ModelStatus.ERROR:
                self.manager._remove(sign)
                raise RuntimeError(f"replica restore failed: {model.error}")
        else:
            threading.Thread(target=_load, daemon=True).start()
        return model

    def delete_model(self, sign: str) -> None:
        m = self.manager._remove(sign)
        if m is None:
            raise KeyError(f"model {sign!r} not found")
        m.status = ModelStatus.DELETED
        m.variables = {}

    def show_models(self) -> List[dict]:
        return [self.manager._models[s].describe()
                for s in self.manager.signs()]

    def show_model(self, sign: str) -> dict:
        m = self.manager.get(sign)
        if m is None:
            raise KeyError(f"model {sign!r} not found")
        return m.describe()

    def show_nodes(self) -> List[dict]:
        """Single-process serving: one node = this process (reference
        show_nodes listed PS server processes)."""
        dev = self.manager.device
        info = {"node_id": 0, "device": dev,
                "models": self.manager.signs(),
                "shutdown_requested": self.shutdown_requested}
        if dev.startswith("cuda") and torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info()
            info["hbm_free_bytes"] = free
            info["hbm_total_bytes"] = total
        return [info]

    def shutdown_node(self, node_id: int = 0) -> None:
        self.shutdown_requested = True


class ServingClient:
    """Minimal HTTP pull client with replica failover.

    The documented HA deployment runs N serving processes behind this
    client (or any load balancer); it is the reference's client-side
    replica machinery — pick_one_replica + retry-until-success
    (EmbeddingPullOperator.cpp:50-58, c_api_test.h:117-121) — over HTTP:
    requests rotate round-robin across endpoints and fail over to the next
    replica on connection error or non-200."""

    def __init__(self, endpoints: List[str], timeout: float = 5.0,
                 attempts_per_endpoint: int = 2):
        if not endpoints:
            raise ValueError("need at least one endpoint")
        self.endpoints = [e.rstrip("/") for e in endpoints]
        self.timeout = timeout
        self.attempts = attempts_per_endpoint * len(self.endpoints)
        self._i = 0

    def _request(self, method: str, path: str, json_body=None):
        import requests

        last: Optional[Exception] = None
        for _ in range(self.attempts):
            ep = self.endpoints[self._i % len(self.endpoints)]
            self._i += 1
            try:
                r = requests.request(method, ep + path, json=json_body,
                                     timeout=self.timeout)
                if r.status_code == 200:
                    return r.json()
                last = RuntimeError(f"{ep}{path}: HTTP {r.status_code} "
                                    f"{r.text[:200]}")
            except Exception as e:  # noqa: BLE001 - any replica failure
                last = e
        raise RuntimeError(f"all {len(self.endpoints)} replicas failed: "
                           f"{last!r}")

    def pull(self, sign: str, variable_id: int, indices) -> List:
        if torch.is_tensor(indices):
            indices = indices.tolist()
        r = self._request("POST",
                          f"/models/{sign}/variables/{variable_id}/pull",
                          {"indices": indices})
        return r["weights"]

    def show_models(self) -> List[dict]:
        return self._request("GET", "/models")


def make_app(controller: Optional[ModelController] = None,
             device: str = "cpu"):
    """REST surface (reference entry/controller.cc routes):
      POST   /models            {"model_uri": ..., "sign"?: ..., "wait"?: bool}
      GET    /models            list
      GET    /models/{sign}     describe
      DELETE /models/{sign}
      GET    /nodes             list nodes
      DELETE /nodes/{id}        request shutdown
      POST   /models/{sign}/variables/{vid}/pull   {"indices": [[...]]}
    The pull route is an addition over the reference (whose serving data path
    went through TF-Serving custom ops): it makes the server usable from any
    HTTP client without TF."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    controller = controller or ModelController(device=device)
    app = FastAPI(title="openembedding_amd serving controller")
    app.state.controller = controller

    class CreateModelReq(BaseModel):
        model_uri: Optional[str] = None
        sign: Optional[str] = None
        wait: bool = True
        # coordinated restore: rebuild from a live replica instead of a
        # dump (requires sign; reference server --restore semantics)
        from_replica: Optional[str] = None

    class PullReq(BaseModel):
        indices: list

    @app.post("/models")
    def create_model(req: CreateModelReq):
        try:
            if req.from_replica:
                if not req.sign:
                    raise HTTPException(status_code=422,
                                        detail="from_replica needs sign")
                m = controller.restore_model_from_replica(
                    req.from_replica, req.sign, wait=req.wait)
            else:
                if not req.model_uri:
                    raise HTTPException(status_code=422,
                                        detail="model_uri required")
                m = controller.create_model(req.model_uri, sign=req.sign,
                                            wait=req.wait)
        except ValueError as e:
            raise HTTPException(status_code=409, detail=str(e))
        except (FileNotFoundError, RuntimeError) as e:
            raise HTTPException(status_code=400, detail=str(e))
        return m.describe()

    @app.get("/models/{sign}/variables/{variable_id}/rows")
    def export_rows(sign: str, variable_id: int, start: int = 0,
                    count: int = 65536):
        """Replica-restore wire: row blocks in export order."""
        m = controller.manager.get(sign)
        if m is None or m.status != ModelStatus.NORMAL:
            raise HTTPException(status_code=404, detail=f"model {sign!r}")
        if variable_id not in m.variables:
            raise HTTPException(status_code=404,
                                detail=f"variable {variable_id}")
        return m.export_block(variable_id, start, count)

    @app.get("/models")
    def list_models():
        return controller.show_models()

    @app.get("/models/{sign}")
    def show_model(sign: str):
        try:
            return controller.show_model(sign)
        except KeyError as e:
            raise HTTPException(status_code=404, detail=str(e))

    @app.delete("/models/{sign}")
    def delete_model(sign: str):
        try:
            controller.delete_model(sign)
        except KeyError as e:
            raise HTTPException(status_code=404, detail=str(e))
        return {"deleted": sign}

    @app.get("/nodes")
    def list_nodes():
        return controller.show_nodes()

    @app.delete("/nodes/{node_id}")
    def shutdown_node(node_id: int):
        controller.shutdown_node(node_id)
        return {"shutdown_requested": node_id}

    @app.get("/metrics")
    def metrics():
        """Prometheus-style exposition of the process accumulators
        (reference server metrics exposer, entry/server.cc:7-12,35-36)."""
        from starlette.responses import PlainTextResponse

        from .utils.metrics import REGISTRY
        lines = ["# TYPE openembedding_metric gauge"]
        for name in REGISTRY.names():
            a = REGISTRY.accumulator(name)
            safe = name.replace(".", "_").replace("-", "_")
            lines.append(f'openembedding_metric{{name="{safe}",stat="count"}}'
                         f' {a.n}')
            lines.append(f'openembedding_metric{{name="{safe}",stat="sum"}}'
                         f' {a.total}')
        lines.append(f'openembedding_models {len(controller.manager.signs())}')
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.post("/models/{sign}/variables/{variable_id}/pull")
    def pull(sign: str, variable_id: int, req: PullReq):
        try:
            var = controller.manager.find_model_variable(sign, variable_id)
        except KeyError as e:
            raise HTTPException(status_code=404, detail=str(e))
        except RuntimeError as e:
            raise HTTPException(status_code=409, detail=str(e))
        idx = torch.tensor(req.indices, dtype=torch.int64)
        out = var.pull_weights(idx)
        return {"weights": out.cpu().tolist()}

    return app


def main():  # pragma: no cover - thin CLI (reference controller daemon)
    import argparse

    import uvicorn

    p = argparse.ArgumentParser(description="openembedding_amd serving "
                                            "controller (REST)")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8010)
    p.add_argument("--device", default=(
        "cuda:0" if torch.cuda.is_available() else "cpu"))
    p.add_argument("--model-uri", action="append", default=[],
                   help="dump URI(s) to load at startup")
    args = p.parse_args()
    controller = ModelController(device=args.device)
    for uri in args.model_uri:
        controller.create_model(uri)
    uvicorn.run(make_app(controller), host=args.host, port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
