"""Numpy interpreter for exported SavedModel GraphDefs (TF-free).

Executes the op set openembedding_amd.export_tf emits, decoding the wire
format with the generic decoder in utils/tfproto.py. An unknown op or
malformed attr fails loudly — it is an INDEPENDENT consumer of the export
(the round-trip tests run predictions through it against the live torch
engine), and doubles as a way to smoke-test an export on machines without
TensorFlow:

    from openembedding_amd.utils.tf_graph_interp import GraphInterp
    g = GraphInterp("/path/to/saved_model")
    ins, outs = g.signature_io()
    logits, probs = g.run({"dense": d, "sparse": s},
                          [outs["logits"], outs["probabilities"]])
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np

from openembedding_amd.utils import tfproto as tp

_NP = {tp.DT_FLOAT: np.float32, tp.DT_DOUBLE: np.float64,
       tp.DT_INT32: np.int32, tp.DT_INT64: np.int64}


def parse_saved_model(path: str):
    """saved_model.pb -> (nodes, signature dict, saver dict, variables)."""
    with open(f"{path}/saved_model.pb", "rb") as f:
        sm = tp.decode_message(f.read())
    assert sm[1][0] == 1, "saved_model_schema_version"
    mg = tp.decode_message(sm[2][0])
    meta_info = tp.decode_message(mg[1][0])
    tags = [t.decode() for t in meta_info.get(4, [])]
    graph = tp.decode_message(mg[2][0])
    saver = tp.decode_message(mg[3][0])
    sigs = {}
    for entry in mg.get(5, []):
        e = tp.decode_message(entry)
        sigs[e[1][0].decode()] = tp.decode_message(e[2][0])
    nodes = [tp.decode_message(n) for n in graph[1][0:]] if 1 in graph else []
    with open(f"{path}/variables/variables.index", "rb") as f:
        idx = f.read()
    with open(f"{path}/variables/variables.data-00000-of-00001", "rb") as f:
        dat = f.read()
    variables = tp.read_bundle(idx, dat)
    return nodes, sigs, saver, variables, tags


def _attrs(node_msg) -> Dict[str, dict]:
    out = {}
    for entry in node_msg.get(5, []):
        e = tp.decode_message(entry)
        out[e[1][0].decode()] = tp.decode_message(e[2][0])
    return out


def _const_value(attr):
    t = tp.decode_message(attr["value"][8][0])
    dtype = t[1][0]
    dims = []
    if 2 in t:
        sh = tp.decode_message(t[2][0])
        for dm in sh.get(2, []):
            dims.append(tp.decode_message(dm)[1][0])
    if dtype == tp.DT_STRING:
        vals = [v for v in t.get(8, [])]
        return np.array(vals, dtype=object).reshape(dims or [len(vals)])
    raw = t.get(4, [b""])[0]
    a = np.frombuffer(raw, dtype=_NP[dtype])
    return a.reshape(dims)


class GraphInterp:
    def __init__(self, path: str):
        (self.nodes, self.sigs, self.saver, self.variables,
         self.tags) = parse_saved_model(path)
        self.by_name = {}
        for n in self.nodes:
            self.by_name[n[1][0].decode()] = n

    def signature_io(self, sig="serving_default"):
        s = self.sigs[sig]
        def side(field):
            out = {}
            for entry in s.get(field, []):
                e = tp.decode_message(entry)
                ti = tp.decode_message(e[2][0])
                out[e[1][0].decode()] = ti[1][0].decode()
            return out
        return side(1), side(2)

    def run(self, feeds: Dict[str, np.ndarray], fetches: List[str]):
        env: Dict[str, np.ndarray] = {}

        def tensor(ref: str):
            name = ref.split(":")[0]
            slot = int(ref.split(":")[1]) if ":" in ref else 0
            v = evaluate(name)
            return v[slot] if isinstance(v, tuple) else v

        def evaluate(name: str):
            if name in env:
                return env[name]
            msg = self.by_name[name]
            op = msg[2][0].decode()
            inputs = [i.decode() for i in msg.get(3, [])]
            data_in = [i for i in inputs if not i.startswith("^")]
            attrs = _attrs(msg)
            env[name] = self._exec(name, op, data_in, attrs, tensor, feeds)
            return env[name]

        return [tensor(f) for f in fetches]

    def _exec(self, name, op, inputs, attrs, tensor, feeds):
        if op == "Placeholder":
            return np.asarray(feeds[name])
        if op == "Const":
            return _const_value(attrs)
        if op == "VariableV2":
            dt, dims, raw = self.variables[name]
            return np.frombuffer(raw, dtype=_NP[dt]).reshape(dims)
        if op == "Identity":
            return tensor(inputs[0])
        x = [tensor(i) for i in inputs]
        if op == "AddV2":
            return x[0] + x[1]
        if op == "Sub":
            return x[0] - x[1]
        if op == "Mul":
            return x[0] * x[1]
        if op == "MatMul":
            return x[0] @ x[1]
        if op == "BiasAdd":
            return x[0] + x[1]
        if op == "Relu":
            return np.maximum(x[0], 0)
        if op == "Sigmoid":
            return 1.0 / (1.0 + np.exp(-x[0]))
        if op == "Reshape":
            return x[0].reshape([int(v) for v in x[1]])
        if op == "Transpose":
            return np.transpose(x[0], [int(v) for v in x[1]])
        if op == "Sum":
            keep = bool(attrs.get("keep_dims", {}).get(5, [0])[0])
            axes = tuple(int(v) for v in np.atleast_1d(x[1]))
            return x[0].sum(axis=axes, keepdims=keep)
        if op == "ConcatV2":
            axis = int(np.atleast_1d(x[-1])[0])
            return np.concatenate(x[:-1], axis=axis)
        if op == "GatherV2":
            axis = int(np.atleast_1d(x[2])[0])
            return np.take(x[0], x[1], axis=axis)
        if op == "Slice":
            begin = [int(v) for v in x[1]]
            size = [int(v) for v in x[2]]
            idx = tuple(
                slice(b, None if s == -1 else b + s)
                for b, s in zip(begin, size))
            return x[0][idx]
        raise NotImplementedError(f"op {op} (node {name})")
