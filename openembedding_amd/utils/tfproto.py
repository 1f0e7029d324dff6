"""Minimal TensorFlow proto / checkpoint-bundle wire codecs (no TF needed).

The north-star export contract is a SavedModel directory TensorFlow
Serving can load (reference save_as_original_model, exb.py:506-547). This
image has no TensorFlow, so the writer speaks the wire formats directly:

- protobuf wire format (encode + decode; decode is used by the round-trip
  tests and by any consumer that wants to inspect an export without TF);
- crc32c (Castagnoli) with TF's rotate-and-add masking (crc32c/crc32c.h);
- the leveldb table format that backs ``variables.index``
  (tensorflow/core/util/tensor_bundle writes a leveldb-compatible SSTable:
  block entries with restarts, per-block type byte + masked crc, footer
  with the 0xdb4775248b80fb57 magic).

Field numbers follow the public tensorflow .proto files (saved_model.proto,
meta_graph.proto, graph.proto, attr_value.proto, tensor.proto,
tensor_shape.proto, saver.proto, tensor_bundle.proto, versions.proto).
"""

from __future__ import annotations

import struct
from typing import Dict, Iterable, List, Tuple, Union

# --------------------------------------------------------------- protobuf

_WT_VARINT = 0
_WT_I64 = 1
_WT_LEN = 2
_WT_I32 = 5


def _varint(v: int) -> bytes:
    if v < 0:
        v += 1 << 64
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def tag(field: int, wt: int) -> bytes:
    return _varint((field << 3) | wt)


def f_varint(field: int, v: int) -> bytes:
    return tag(field, _WT_VARINT) + _varint(int(v))


def f_bool(field: int, v: bool) -> bytes:
    return f_varint(field, 1 if v else 0)


def f_bytes(field: int, v: Union[bytes, str]) -> bytes:
    if isinstance(v, str):
        v = v.encode("utf-8")
    return tag(field, _WT_LEN) + _varint(len(v)) + v


def f_msg(field: int, body: bytes) -> bytes:
    return f_bytes(field, body)


def f_float(field: int, v: float) -> bytes:
    return tag(field, _WT_I32) + struct.pack("<f", v)


def f_fixed32(field: int, v: int) -> bytes:
    return tag(field, _WT_I32) + struct.pack("<I", v & 0xFFFFFFFF)


def f_map_entry(field: int, key: Union[str, bytes], value_msg: bytes,
                key_field: int = 1, value_field: int = 2) -> bytes:
    """map<string, Message> entry: submessage {1: key, 2: value}."""
    return f_msg(field, f_bytes(key_field, key) + f_msg(value_field,
                                                        value_msg))


def decode_message(data: bytes) -> Dict[int, List]:
    """Wire-level decode: field -> list of raw values (int for varint/fixed,
    bytes for length-delimited). Nested messages decode lazily via another
    decode_message call on the bytes."""
    out: Dict[int, List] = {}
    i = 0
    n = len(data)
    while i < n:
        key, i = _read_varint(data, i)
        field, wt = key >> 3, key & 7
        if wt == _WT_VARINT:
            v, i = _read_varint(data, i)
        elif wt == _WT_I64:
            v = struct.unpack_from("<Q", data, i)[0]
            i += 8
        elif wt == _WT_LEN:
            ln, i = _read_varint(data, i)
            v = data[i:i + ln]
            i += ln
        elif wt == _WT_I32:
            v = struct.unpack_from("<I", data, i)[0]
            i += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")
        out.setdefault(field, []).append(v)
    return out


def _read_varint(data: bytes, i: int) -> Tuple[int, int]:
    shift = 0
    v = 0
    while True:
        b = data[i]
        i += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, i
        shift += 7


# ------------------------------------------------------------------ crc32c

_CRC_TABLE = []


def _crc_init():
    poly = 0x82F63B78  # Castagnoli, reflected
    for n in range(256):
        c = n
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        _CRC_TABLE.append(c)


_crc_init()


def crc32c(data: bytes, crc: int = 0) -> int:
    c = crc ^ 0xFFFFFFFF
    for b in data:
        c = _CRC_TABLE[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def masked_crc32c(data: bytes) -> int:
    """TF's crc mask (crc32c.h): rotate right 15 and add a constant."""
    crc = crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ----------------------------------------------------- leveldb table writer

_TABLE_MAGIC = 0xDB4775248B80FB57


def _block(entries: Iterable[Tuple[bytes, bytes]]) -> bytes:
    """One leveldb block, restart point at every entry (shared prefix 0)."""
    buf = bytearray()
    restarts = []
    for key, value in entries:
        restarts.append(len(buf))
        buf += _varint(0)            # shared
        buf += _varint(len(key))     # non-shared
        buf += _varint(len(value))
        buf += key
        buf += value
    for r in restarts:
        buf += struct.pack("<I", r)
    buf += struct.pack("<I", max(1, len(restarts)) if restarts else 0)
    if not restarts:
        # empty block still records one restart array slot
        buf = bytearray(struct.pack("<II", 0, 1))
    return bytes(buf)


def _block_handle(offset: int, size: int) -> bytes:
    return _varint(offset) + _varint(size)


def write_table(entries: List[Tuple[bytes, bytes]]) -> bytes:
    """A one-data-block leveldb table (sorted small key set — exactly the
    shape of a variables.index file)."""
    entries = sorted(entries)
    out = bytearray()

    def emit_block(b: bytes) -> Tuple[int, int]:
        off = len(out)
        out.extend(b)
        out.append(0)  # compression: none
        out.extend(struct.pack("<I", masked_crc32c(b + b"\x00")))
        return off, len(b)

    data_off, data_sz = emit_block(_block(entries))
    meta_off, meta_sz = emit_block(_block([]))
    last_key = entries[-1][0] if entries else b""
    index_entries = [(last_key + b"\x00", _block_handle(data_off, data_sz))]
    idx_off, idx_sz = emit_block(_block(index_entries))

    footer = (_block_handle(meta_off, meta_sz)
              + _block_handle(idx_off, idx_sz))
    footer += b"\x00" * (40 - len(footer))
    footer += struct.pack("<Q", _TABLE_MAGIC)
    out.extend(footer)
    return bytes(out)


def read_table(data: bytes) -> List[Tuple[bytes, bytes]]:
    """Parse a leveldb table back to (key, value) pairs (test-side reader —
    independent check that the writer produced a conformant file)."""
    magic = struct.unpack_from("<Q", data, len(data) - 8)[0]
    if magic != _TABLE_MAGIC:
        raise ValueError("bad table magic")
    footer = data[len(data) - 48:len(data) - 8]
    i = 0
    _meta_off, i = _read_varint(footer, i)
    _meta_sz, i = _read_varint(footer, i)
    idx_off, i = _read_varint(footer, i)
    idx_sz, i = _read_varint(footer, i)
    idx = _parse_block(data[idx_off:idx_off + idx_sz])
    pairs: List[Tuple[bytes, bytes]] = []
    for _key, handle in idx:
        j = 0
        off, j = _read_varint(handle, j)
        sz, j = _read_varint(handle, j)
        # verify the stored block crc like a real reader would
        blk = data[off:off + sz]
        ctype = data[off + sz]
        stored = struct.unpack_from("<I", data, off + sz + 1)[0]
        if masked_crc32c(blk + bytes([ctype])) != stored:
            raise ValueError("block crc mismatch")
        pairs.extend(_parse_block(blk))
    return pairs


def _parse_block(b: bytes) -> List[Tuple[bytes, bytes]]:
    n_restarts = struct.unpack_from("<I", b, len(b) - 4)[0]
    end = len(b) - 4 - 4 * n_restarts
    i = 0
    out = []
    prev_key = b""
    while i < end:
        shared, i = _read_varint(b, i)
        non_shared, i = _read_varint(b, i)
        vlen, i = _read_varint(b, i)
        key = prev_key[:shared] + b[i:i + non_shared]
        i += non_shared
        val = b[i:i + vlen]
        i += vlen
        out.append((key, val))
        prev_key = key
    return out


# ------------------------------------------------- TF message constructors

# types.proto
DT_FLOAT = 1
DT_DOUBLE = 2
DT_INT32 = 3
DT_STRING = 7
DT_INT64 = 9
DT_BOOL = 10

_DT_NP = {DT_FLOAT: "<f4", DT_DOUBLE: "<f8", DT_INT32: "<i4",
          DT_INT64: "<i8"}


def dt_numpy(dtype: int):
    import numpy as np
    return np.dtype(_DT_NP[dtype])


def shape_proto(dims) -> bytes:
    """TensorShapeProto: repeated Dim dim = 2 {int64 size = 1}."""
    out = b""
    for d in dims:
        out += f_msg(2, f_varint(1, int(d)))
    return out


def tensor_proto(dtype: int, dims, content: bytes) -> bytes:
    """TensorProto with tensor_content (field 4)."""
    return (f_varint(1, dtype) + f_msg(2, shape_proto(dims))
            + f_bytes(4, content))


def tensor_proto_strings(values: List[bytes]) -> bytes:
    body = f_varint(1, DT_STRING) + f_msg(2, shape_proto([len(values)]))
    for v in values:
        body += f_bytes(8, v)   # string_val
    return body


def tensor_proto_scalar_string(value: bytes) -> bytes:
    return f_varint(1, DT_STRING) + f_msg(2, shape_proto([])) \
        + f_bytes(8, value)


def attr_type(dt: int) -> bytes:
    return f_varint(6, dt)              # AttrValue.type


def attr_int(v: int) -> bytes:
    return f_varint(3, v)               # AttrValue.i


def attr_bool(v: bool) -> bytes:
    return f_bool(5, v)                 # AttrValue.b


def attr_shape(dims) -> bytes:
    return f_msg(7, shape_proto(dims))  # AttrValue.shape


def attr_tensor(tp: bytes) -> bytes:
    return f_msg(8, tp)                 # AttrValue.tensor

def attr_list_types(dts) -> bytes:
    body = b"".join(_varint(d) for d in dts)
    return f_msg(1, tag(6, _WT_LEN) + _varint(len(body)) + body)


def node(name: str, op: str, inputs: List[str] = (),
         attrs: Dict[str, bytes] = None) -> bytes:
    """NodeDef: name=1, op=2, input=3, attr=5 (map<string, AttrValue>)."""
    body = f_bytes(1, name) + f_bytes(2, op)
    for i in inputs:
        body += f_bytes(3, i)
    for k, v in (attrs or {}).items():
        body += f_map_entry(5, k, v)
    return body


def graph_def(nodes: List[bytes], producer: int = 1395) -> bytes:
    """GraphDef: node=1, versions=4 (producer=1)."""
    body = b"".join(f_msg(1, n) for n in nodes)
    body += f_msg(4, f_varint(1, producer) + f_varint(2, 12))
    return body


def tensor_info(name: str, dtype: int, dims) -> bytes:
    return f_bytes(1, name) + f_varint(2, dtype) + f_msg(3,
                                                         shape_proto(dims))


def signature_def(inputs: Dict[str, bytes], outputs: Dict[str, bytes],
                  method: str = "tensorflow/serving/predict") -> bytes:
    body = b""
    for k, v in inputs.items():
        body += f_map_entry(1, k, v)
    for k, v in outputs.items():
        body += f_map_entry(2, k, v)
    body += f_bytes(3, method)
    return body


def saver_def(filename_tensor: str, restore_op: str,
              save_tensor: str) -> bytes:
    """SaverDef: filename_tensor_name=1, save_tensor_name=2,
    restore_op_name=3, max_to_keep=4, version=7 (V2=2)."""
    return (f_bytes(1, filename_tensor) + f_bytes(2, save_tensor)
            + f_bytes(3, restore_op) + f_varint(4, 5) + f_varint(7, 2))


def meta_graph(graph: bytes, signatures: Dict[str, bytes],
               saver: bytes, tags: List[str] = ("serve",)) -> bytes:
    meta_info = b"".join(f_bytes(4, t) for t in tags)
    meta_info += f_bytes(5, "2.12.0-oeamd")  # tensorflow_version (cosmetic)
    body = f_msg(1, meta_info) + f_msg(2, graph) + f_msg(3, saver)
    for k, v in signatures.items():
        body += f_map_entry(5, k, v)
    return body


def saved_model(meta_graphs: List[bytes]) -> bytes:
    body = f_varint(1, 1)   # saved_model_schema_version
    for mg in meta_graphs:
        body += f_msg(2, mg)
    return body


# -------------------------------------------------- variables bundle writer

def bundle_header(num_shards: int = 1) -> bytes:
    """BundleHeaderProto: num_shards=1, endianness=2 (LITTLE=0),
    version=3 (VersionDef producer=1)."""
    return f_varint(1, num_shards) + f_msg(3, f_varint(1, 1))


def bundle_entry(dtype: int, dims, shard_id: int, offset: int, size: int,
                 crc: int) -> bytes:
    """BundleEntryProto: dtype=1 shape=2 shard_id=3 offset=4 size=5
    crc32c=6 (fixed32)."""
    body = f_varint(1, dtype) + f_msg(2, shape_proto(dims))
    if shard_id:
        body += f_varint(3, shard_id)
    if offset:
        body += f_varint(4, offset)
    body += f_varint(5, size) + f_fixed32(6, crc)
    return body


def write_bundle(tensors: Dict[str, Tuple[int, List[int], bytes]]
                 ) -> Tuple[bytes, bytes]:
    """tensors: name -> (dtype, dims, raw_bytes). Returns
    (variables.index bytes, variables.data-00000-of-00001 bytes)."""
    data = bytearray()
    entries: List[Tuple[bytes, bytes]] = []
    for name in sorted(tensors):
        dtype, dims, raw = tensors[name]
        off = len(data)
        data.extend(raw)
        entries.append((name.encode(),
                        bundle_entry(dtype, dims, 0, off, len(raw),
                                     masked_crc32c(raw))))
    index_entries = [(b"", bundle_header())] + entries
    return write_table(index_entries), bytes(data)


def read_bundle(index_bytes: bytes, data_bytes: bytes):
    """Test-side reader: name -> (dtype, dims, raw)."""
    out = {}
    for key, val in read_table(index_bytes):
        if key == b"":
            continue
        msg = decode_message(val)
        dtype = msg.get(1, [DT_FLOAT])[0]
        dims = []
        if 2 in msg:
            shape = decode_message(msg[2][0])
            for dim_msg in shape.get(2, []):
                dims.append(decode_message(dim_msg).get(1, [0])[0])
        offset = msg.get(4, [0])[0]
        size = msg.get(5, [0])[0]
        crc = msg.get(6, [0])[0]
        raw = data_bytes[offset:offset + size]
        if masked_crc32c(raw) != crc:
            raise ValueError(f"crc mismatch for {key!r}")
        out[key.decode()] = (dtype, dims, raw)
    return out
