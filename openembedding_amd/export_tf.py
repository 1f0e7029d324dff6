"""TensorFlow-Serving-loadable SavedModel export (no TF dependency).

The reference's ``save_as_original_model`` (exb.py:506-547) clones the
Keras graph, materializes every PS row into a vanilla Embedding, and saves
a standard SavedModel that TensorFlow Serving loads with no OpenEmbedding
runtime. This module reproduces that contract from the torch side: it
materializes the engine's rows and writes a TF1-style SavedModel directory

    path/saved_model.pb                        (MetaGraphDef, tags=[serve])
    path/variables/variables.index             (TensorBundle / leveldb table)
    path/variables/variables.data-00000-of-00001

by emitting the protobuf wire format directly (utils/tfproto.py — this
image has no TF). The graph is plain inference ops (Placeholder, GatherV2,
MatMul, BiasAdd, Relu, Mul, Sum, Sigmoid, ...) plus the standard
RestoreV2/SaveV2 saver subgraph that SavedModel loaders drive through
SaverDef; signature ``serving_default`` takes (dense [B,13] float32,
sparse [B,26] int64 per-field ids) and returns logits + probabilities.

Model families covered: the zoo (LR, WDL, DeepFM, xDeepFM) — the same
coverage the reference's benchmark exports exercised. Correctness is
checked TF-free by tests/test_export_tf.py: an independent decoder parses
the wire format back and a numpy interpreter executes the exported graph
against the live torch model.
"""

from __future__ import annotations

import os
from typing import Dict, List, Tuple

import numpy as np
import torch

from .utils import tfproto as tp


class GraphBuilder:
    def __init__(self):
        self.nodes: List[bytes] = []
        self.variables: Dict[str, Tuple[int, List[int], bytes]] = {}

    def add(self, name: str, op: str, inputs=(), **attrs) -> str:
        self.nodes.append(tp.node(name, op, list(inputs), attrs))
        return name

    # -- constants / variables ------------------------------------------

    def const(self, name: str, arr: np.ndarray) -> str:
        dt = _np_dt(arr.dtype)
        self.add(name, "Const",
                 dtype=tp.attr_type(dt),
                 value=tp.attr_tensor(tp.tensor_proto(
                     dt, arr.shape, arr.astype(arr.dtype).tobytes())))
        return name

    def const_i32(self, name: str, values) -> str:
        return self.const(name, np.asarray(values, dtype=np.int32))

    def variable(self, name: str, arr: np.ndarray) -> str:
        """VariableV2 + read Identity; registers the tensor for the bundle
        (checkpoint key == variable node name, TF1 convention)."""
        dt = _np_dt(arr.dtype)
        self.add(name, "VariableV2",
                 dtype=tp.attr_type(dt),
                 shape=tp.attr_shape(arr.shape),
                 container=_attr_s(b""),
                 shared_name=_attr_s(b""))
        self.add(name + "/read", "Identity", [name], T=tp.attr_type(dt))
        self.variables[name] = (dt, list(arr.shape),
                                np.ascontiguousarray(arr).tobytes())
        return name + "/read"

    # -- math helpers ----------------------------------------------------

    def matmul(self, name, a, b):
        return self.add(name, "MatMul", [a, b], T=tp.attr_type(tp.DT_FLOAT),
                        transpose_a=tp.attr_bool(False),
                        transpose_b=tp.attr_bool(False))

    def bias_add(self, name, x, b):
        return self.add(name, "BiasAdd", [x, b],
                        T=tp.attr_type(tp.DT_FLOAT))

    def relu(self, name, x):
        return self.add(name, "Relu", [x], T=tp.attr_type(tp.DT_FLOAT))

    def add_f(self, name, a, b):
        return self.add(name, "AddV2", [a, b],
                        T=tp.attr_type(tp.DT_FLOAT))

    def mul(self, name, a, b):
        return self.add(name, "Mul", [a, b], T=tp.attr_type(tp.DT_FLOAT))

    def sub(self, name, a, b):
        return self.add(name, "Sub", [a, b], T=tp.attr_type(tp.DT_FLOAT))

    def reshape(self, name, x, shape):
        s = self.const_i32(name + "/shape", shape)
        return self.add(name, "Reshape", [x, s],
                        T=tp.attr_type(tp.DT_FLOAT),
                        Tshape=tp.attr_type(tp.DT_INT32))

    def reduce_sum(self, name, x, axes, keep_dims=False):
        a = self.const_i32(name + "/axes", axes)
        return self.add(name, "Sum", [x, a], T=tp.attr_type(tp.DT_FLOAT),
                        Tidx=tp.attr_type(tp.DT_INT32),
                        keep_dims=tp.attr_bool(keep_dims))

    def transpose(self, name, x, perm):
        p = self.const_i32(name + "/perm", perm)
        return self.add(name, "Transpose", [x, p],
                        T=tp.attr_type(tp.DT_FLOAT),
                        Tperm=tp.attr_type(tp.DT_INT32))

    def concat(self, name, values, axis):
        a = self.const_i32(name + "/axis", [axis])
        return self.add(name, "ConcatV2", list(values) + [a],
                        N=tp.attr_int(len(values)),
                        T=tp.attr_type(tp.DT_FLOAT),
                        Tidx=tp.attr_type(tp.DT_INT32))

    def slice(self, name, x, begin, size):
        b = self.const_i32(name + "/begin", begin)
        s = self.const_i32(name + "/size", size)
        return self.add(name, "Slice", [x, b, s],
                        T=tp.attr_type(tp.DT_FLOAT),
                        Index=tp.attr_type(tp.DT_INT32))

    def mlp(self, prefix, x, torch_seq):
        """Linear/ReLU stack from a torch nn.Sequential."""
        import torch.nn as nn
        h = x
        li = 0
        for mod in torch_seq:
            if isinstance(mod, nn.Linear):
                w = self.variable(f"{prefix}/w{li}",
                                  _np(mod.weight).T.copy())
                b = self.variable(f"{prefix}/b{li}", _np(mod.bias))
                h = self.matmul(f"{prefix}/mm{li}", h, w)
                h = self.bias_add(f"{prefix}/ba{li}", h, b)
                li += 1
            elif isinstance(mod, nn.ReLU):
                h = self.relu(f"{prefix}/relu{li}", h)
            else:
                raise NotImplementedError(f"layer {type(mod).__name__}")
        return h

    # -- saver subgraph --------------------------------------------------

    def build_saver(self):
        names = sorted(self.variables)
        self.add("save/Const", "Const",
                 dtype=tp.attr_type(tp.DT_STRING),
                 value=tp.attr_tensor(
                     tp.tensor_proto_scalar_string(b"model")))
        self.const_strings("save/RestoreV2/tensor_names",
                           [n.encode() for n in names])
        self.const_strings("save/RestoreV2/shape_and_slices",
                           [b"" for _ in names])
        dtypes = [self.variables[n][0] for n in names]
        self.add("save/RestoreV2", "RestoreV2",
                 ["save/Const", "save/RestoreV2/tensor_names",
                  "save/RestoreV2/shape_and_slices"],
                 dtypes=tp.attr_list_types(dtypes))
        assigns = []
        for i, n in enumerate(names):
            src = "save/RestoreV2" if i == 0 else f"save/RestoreV2:{i}"
            a = self.add(f"save/Assign_{i}", "Assign", [n, src],
                         T=tp.attr_type(self.variables[n][0]),
                         use_locking=tp.attr_bool(True),
                         validate_shape=tp.attr_bool(True))
            assigns.append(a)
        self.add("save/restore_all", "NoOp",
                 [f"^{a}" for a in assigns])
        self.add("save/SaveV2", "SaveV2",
                 ["save/Const", "save/RestoreV2/tensor_names",
                  "save/RestoreV2/shape_and_slices"]
                 + [n + "/read" for n in names],
                 dtypes=tp.attr_list_types(dtypes))
        self.add("save/control_dependency", "Identity",
                 ["save/Const", "^save/SaveV2"],
                 T=tp.attr_type(tp.DT_STRING))

    def const_strings(self, name, values: List[bytes]):
        self.add(name, "Const", dtype=tp.attr_type(tp.DT_STRING),
                 value=tp.attr_tensor(tp.tensor_proto_strings(values)))
        return name


def _attr_s(v: bytes) -> bytes:
    return tp.f_bytes(2, v)     # AttrValue.s


def _np_dt(dtype) -> int:
    dtype = np.dtype(dtype)
    if dtype == np.float32:
        return tp.DT_FLOAT
    if dtype == np.float64:
        return tp.DT_DOUBLE
    if dtype == np.int32:
        return tp.DT_INT32
    if dtype == np.int64:
        return tp.DT_INT64
    raise NotImplementedError(str(dtype))


def _np(t: torch.Tensor) -> np.ndarray:
    return t.detach().to(torch.float32).cpu().numpy()


# ---------------------------------------------------------------- models

def _materialize_embedding(model) -> np.ndarray:
    """All rows [0, total_vocab) of the CombinedEmbedding, instantiated
    like the reference bulk pull (missing rows get their initializer
    values, exb.py:529-538)."""
    from .torch import sparse_read_as_dense
    emb = model.embedding
    total = int(emb.field_offsets[-1].item()) + int(model.field_vocabs[-1])
    rows = sparse_read_as_dense(emb.variable, total)
    return _np(rows)


def _common_inputs(g: GraphBuilder, model):
    from .models.criteo import N_DENSE
    n_fields = model.n_fields
    dim1 = model.dim + 1
    g.add("dense", "Placeholder", dtype=tp.attr_type(tp.DT_FLOAT),
          shape=tp.attr_shape([-1, N_DENSE]))
    g.add("sparse", "Placeholder", dtype=tp.attr_type(tp.DT_INT64),
          shape=tp.attr_shape([-1, n_fields]))
    offsets = g.const("field_offsets",
                      _np(model.embedding.field_offsets).astype(np.int64))
    g.add("keys", "AddV2", ["sparse", offsets],
          T=tp.attr_type(tp.DT_INT64))
    table = g.variable("embedding", _materialize_embedding(model))
    axis = g.const_i32("gather_axis", [0])
    g.add("e_all", "GatherV2", [table, "keys", axis],
          Tparams=tp.attr_type(tp.DT_FLOAT),
          Tindices=tp.attr_type(tp.DT_INT64),
          Taxis=tp.attr_type(tp.DT_INT32),
          batch_dims=tp.attr_int(0))
    # e [B,F,dim], lin [B,F]
    dim = model.dim
    e = None
    if dim > 0:
        e = g.slice("e", "e_all", [0, 0, 0], [-1, -1, dim])
    lin3 = g.slice("lin3", "e_all", [0, 0, dim], [-1, -1, 1])
    lin = g.reshape("lin", lin3, [-1, n_fields])
    return e, lin, dim1


def _first_order(g: GraphBuilder, model, lin) -> str:
    wd = g.variable("dense_linear/w", _np(model.dense_linear.weight).T.copy())
    bd = g.variable("dense_linear/b", _np(model.dense_linear.bias))
    dl = g.matmul("dense_linear/mm", "dense", wd)
    dl = g.bias_add("dense_linear/out", dl, bd)      # [B,1]
    lsum = g.reduce_sum("lin_sum", lin, [1], keep_dims=True)  # [B,1]
    return g.add_f("first_order", dl, lsum)          # [B,1]


def _deep(g: GraphBuilder, model, e) -> str:
    from .models.criteo import N_DENSE
    flat = g.reshape("e_flat", e, [-1, model.n_fields * model.dim])
    deep_in = g.concat("deep_in", [flat, "dense"], 1)
    return g.mlp("dnn", deep_in, model.dnn)          # [B,1]


def _fm(g: GraphBuilder, model, e) -> str:
    s = g.reduce_sum("fm/s", e, [1])                         # [B,d]
    ss = g.mul("fm/ss", s, s)
    ee = g.mul("fm/ee", e, e)
    se = g.reduce_sum("fm/se", ee, [1])                      # [B,d]
    diff = g.sub("fm/diff", ss, se)
    half = g.const("fm/half", np.asarray(0.5, dtype=np.float32))
    scaled = g.mul("fm/scaled", diff, half)
    return g.reduce_sum("fm/out", scaled, [1], keep_dims=True)   # [B,1]


def _cin(g: GraphBuilder, model) -> str:
    cin = model.cin
    F = model.n_fields
    d = model.dim
    pooled = []
    xk = "e"
    hk = F
    for li, w in enumerate(cin.weights):
        O = w.shape[0]
        x0r = g.reshape(f"cin{li}/x0r", "e", [-1, F, 1, d])
        xkr = g.reshape(f"cin{li}/xkr", xk, [-1, 1, hk, d])
        m = g.mul(f"cin{li}/outer", x0r, xkr)        # [B,F,hk,d]
        z = g.reshape(f"cin{li}/z", m, [-1, F * hk, d])
        zt = g.transpose(f"cin{li}/zt", z, [0, 2, 1])     # [B,d,FH]
        zz = g.reshape(f"cin{li}/zz", zt, [-1, F * hk])
        wt = g.variable(f"cin{li}/w", _np(w).T.copy())    # [FH,O]
        y = g.matmul(f"cin{li}/mm", zz, wt)               # [B*d,O]
        y2 = g.reshape(f"cin{li}/y2", y, [-1, d, O])
        y3 = g.transpose(f"cin{li}/y3", y2, [0, 2, 1])    # [B,O,d]
        xk = g.relu(f"cin{li}/relu", y3)
        pooled.append(g.reduce_sum(f"cin{li}/pool", xk, [2]))  # [B,O]
        hk = O
    cat = g.concat("cin/cat", pooled, 1)
    wf = g.variable("cin/fc_w", _np(cin.fc.weight).T.copy())
    bf = g.variable("cin/fc_b", _np(cin.fc.bias))
    out = g.matmul("cin/fc_mm", cat, wf)
    return g.bias_add("cin/out", out, bf)            # [B,1]


def build_graph(model) -> GraphBuilder:
    """TF graph for one model of the zoo, mirroring models/ctr.py math."""
    from .models import ctr

    g = GraphBuilder()
    e, lin, _ = _common_inputs(g, model)
    logits = _first_order(g, model, lin)
    if isinstance(model, ctr.DeepFM):
        logits = g.add_f("logits_fm", logits, _fm(g, model, e))
        logits = g.add_f("logits2", logits, _deep(g, model, e))
    elif isinstance(model, ctr.xDeepFM):
        logits = g.add_f("logits_cin", logits, _cin(g, model))
        logits = g.add_f("logits2", logits, _deep(g, model, e))
    elif isinstance(model, ctr.WDL):
        logits = g.add_f("logits2", logits, _deep(g, model, e))
    elif isinstance(model, ctr.LR):
        pass
    else:
        raise NotImplementedError(type(model).__name__)
    g.reshape("logits", logits, [-1])
    g.add("probabilities", "Sigmoid", ["logits"],
          T=tp.attr_type(tp.DT_FLOAT))
    g.build_saver()
    return g


def export_saved_model(model, path: str) -> None:
    """Write the SavedModel directory for ``model`` (rank 0 writes;
    materialization is collective — call on every rank)."""
    from .context import get_context
    from .models.criteo import N_DENSE

    g = build_graph(model)
    graph = tp.graph_def(g.nodes)
    sig = tp.signature_def(
        inputs={
            "dense": tp.tensor_info("dense:0", tp.DT_FLOAT, [-1, N_DENSE]),
            "sparse": tp.tensor_info("sparse:0", tp.DT_INT64,
                                     [-1, model.n_fields]),
        },
        outputs={
            "logits": tp.tensor_info("logits:0", tp.DT_FLOAT, [-1]),
            "probabilities": tp.tensor_info("probabilities:0", tp.DT_FLOAT,
                                            [-1]),
        })
    saver = tp.saver_def("save/Const:0", "save/restore_all",
                         "save/control_dependency:0")
    mg = tp.meta_graph(graph, {"serving_default": sig}, saver)
    sm = tp.saved_model([mg])
    if get_context().rank != 0:
        return
    os.makedirs(os.path.join(path, "variables"), exist_ok=True)
    with open(os.path.join(path, "saved_model.pb"), "wb") as f:
        f.write(sm)
    index, data = tp.write_bundle(g.variables)
    with open(os.path.join(path, "variables", "variables.index"),
              "wb") as f:
        f.write(index)
    with open(os.path.join(path, "variables",
                           "variables.data-00000-of-00001"), "wb") as f:
        f.write(data)
