"""Criteo-format data pipeline.

The reference trained from csv/tfrecord datasets (its benchmark grid ran
{tfrecord, csv} inputs; laboratory/benchmark/benchmark.py:35-68, plus
examples/criteo_preprocess.py for the raw→csv step). This module is the
torch-side equivalent for the raw Criteo Kaggle/Terabyte TSV format:

    label \t I1..I13 (ints, may be empty) \t C1..C26 (hex tokens, may be empty)

- ``CriteoTSV`` parses a file into ready batches: dense int features are
  log1p-normalized fp32 (the standard Criteo recipe, same as the
  reference's preprocess), categorical tokens become int64 keys either by
  per-field modulo into bounded vocabularies (array-mode tables) or by a
  64-bit field-salted hash over the raw token (hash-mode tables,
  ``hash_mode=True`` — any key, lazily-created rows).
- ``BackgroundLoader`` wraps any batch iterator with a producer thread +
  bounded queue so parsing overlaps training; combine with
  ``openembedding_amd.torch.pulling`` for the full reference pipeline
  (dataset thread → prefetched embedding pull → train step).
"""

from __future__ import annotations

import queue
import threading
from typing import Iterable, Iterator, List, Optional, Tuple

import torch

from .models.criteo import CRITEO_FIELD_VOCABS, N_DENSE

N_SPARSE = 26

_MIX = 0x9E3779B97F4A7C15


def _hash_token(field: int, token: str) -> int:
    """Field-salted 64-bit stable hash of a raw categorical token
    (splitmix64-style finalizer over python's string hash would not be
    stable across runs — PYTHONHASHSEED — so mix an explicit FNV-1a)."""
    h = 0xCBF29CE484222325
    for b in token.encode():
        h = ((h ^ b) * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    z = (h + (field + 1) * _MIX) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    z ^= z >> 31
    return z & 0x7FFFFFFFFFFFFFFF  # keep int64-positive


class CriteoTSV:
    """Iterate (dense [B,13] fp32, sparse [B,26] int64, labels [B] fp32)
    batches out of a Criteo-format TSV file.

    hash_mode=False (default): key = token-hash % field_vocabs[f] — bounded
    per-field ids for array tables (``CombinedEmbedding(field_vocabs, …)``).
    hash_mode=True: key = raw 63-bit token hash — feed to hash-mode tables
    (``Embedding(-1, …)``); missing tokens map to the field's key 0 /
    field-salt hash of "".
    """

    def __init__(self, path: str, batch_size: int,
                 field_vocabs: Optional[List[int]] = None,
                 hash_mode: bool = False,
                 vocab_maps: Optional[List[dict]] = None,
                 drop_last: bool = False):
        self.path = path
        self.batch_size = int(batch_size)
        self.field_vocabs = list(field_vocabs or CRITEO_FIELD_VOCABS)
        if len(self.field_vocabs) != N_SPARSE:
            raise ValueError(f"need {N_SPARSE} field vocabs")
        self.hash_mode = hash_mode
        self.vocab_maps = vocab_maps  # from build_vocabs: token->id, 0 = OOV
        self.drop_last = drop_last

    def _emit(self, rows) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        labels = torch.tensor([r[0] for r in rows], dtype=torch.float32)
        dense = torch.tensor([r[1] for r in rows], dtype=torch.float32)
        sparse = torch.tensor([r[2] for r in rows], dtype=torch.int64)
        return dense, sparse, labels

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, ...]]:
        rows = []
        with open(self.path) as f:
            for line in f:
                parts = line.rstrip("\n").split("\t")
                if len(parts) != 1 + N_DENSE + N_SPARSE:
                    continue  # malformed line: skip (reference preprocess
                    #           tolerated ragged raw data the same way)
                label = float(parts[0])
                dense = [torch.log1p(torch.tensor(max(float(v), 0.0))).item()
                         if v else 0.0 for v in parts[1:1 + N_DENSE]]
                cats = []
                for fidx, tok in enumerate(parts[1 + N_DENSE:]):
                    if self.vocab_maps is not None:
                        cats.append(self.vocab_maps[fidx].get(tok, 0))
                    elif self.hash_mode:
                        cats.append(_hash_token(fidx, tok))
                    else:
                        cats.append(_hash_token(fidx, tok)
                                    % self.field_vocabs[fidx])
                rows.append((label, dense, cats))
                if len(rows) == self.batch_size:
                    yield self._emit(rows)
                    rows = []
        if rows and not self.drop_last:
            yield self._emit(rows)


def build_vocabs(path: str, min_count: int = 2
                 ) -> Tuple[List[int], List[dict]]:
    """One scan over a Criteo TSV: per-field token→id maps with a
    frequency threshold (tokens seen < min_count collapse into the
    field's OOV id 0 — the reference's criteo_preprocess.py recipe).

    Returns (field_vocabs, field_maps): `field_vocabs[f]` is the table
    size to give `CombinedEmbedding`, `field_maps[f][token] -> id` for
    encoding; feed both back via ``CriteoTSV(..., vocab_maps=field_maps)``.
    """
    from collections import Counter
    counters = [Counter() for _ in range(N_SPARSE)]
    with open(path) as f:
        for line in f:
            parts = line.rstrip("\n").split("\t")
            if len(parts) != 1 + N_DENSE + N_SPARSE:
                continue
            for fidx, tok in enumerate(parts[1 + N_DENSE:]):
                counters[fidx][tok] += 1
    maps: List[dict] = []
    sizes: List[int] = []
    for c in counters:
        m = {}
        nxt = 1                      # 0 reserved for OOV / rare
        for tok, cnt in sorted(c.items()):
            if cnt >= min_count:
                m[tok] = nxt
                nxt += 1
        maps.append(m)
        sizes.append(max(nxt, 2))
    return sizes, maps


class CriteoTFRecord:
    """Iterate batches out of a TFRecord file of tf.train.Example records —
    the reference benchmark grid's other input format
    (laboratory/benchmark/benchmark.py:35-68 ran {tfrecord, csv}).

    Expected features per Example (the standard Criteo tfrecord layout):
      "label"      float_list or int64_list, 1 value
      "I1".."I13"  float_list (missing -> 0.0)
      "C1".."C26"  int64_list (ready ids) OR bytes_list (raw tokens,
                   hashed/bucketed exactly like CriteoTSV)

    The record framing (length + masked crc32c + payload + masked crc32c)
    and the Example proto are decoded with utils/tfproto.py — both crcs are
    verified; a corrupt record raises."""

    def __init__(self, path: str, batch_size: int,
                 field_vocabs: Optional[List[int]] = None,
                 hash_mode: bool = False, drop_last: bool = False):
        self.path = path
        self.batch_size = int(batch_size)
        self.field_vocabs = list(field_vocabs or CRITEO_FIELD_VOCABS)
        self.hash_mode = hash_mode
        self.drop_last = drop_last

    def _records(self) -> Iterator[bytes]:
        import struct

        from .utils.tfproto import masked_crc32c
        with open(self.path, "rb") as f:
            while True:
                hdr = f.read(12)
                if len(hdr) < 12:
                    return
                (length,) = struct.unpack("<Q", hdr[:8])
                (lcrc,) = struct.unpack("<I", hdr[8:])
                if masked_crc32c(hdr[:8]) != lcrc:
                    raise ValueError("tfrecord length crc mismatch")
                data = f.read(length)
                (dcrc,) = struct.unpack("<I", f.read(4))
                if masked_crc32c(data) != dcrc:
                    raise ValueError("tfrecord data crc mismatch")
                yield data

    def _features(self, record: bytes) -> dict:
        from .utils.tfproto import decode_message
        ex = decode_message(record)
        feats = {}
        features = decode_message(ex[1][0])
        for entry in features.get(1, []):
            e = decode_message(entry)
            name = e[1][0].decode()
            feature = decode_message(e[2][0])
            if 1 in feature:        # BytesList
                bl = decode_message(feature[1][0])
                feats[name] = [v for v in bl.get(1, [])]
            elif 2 in feature:      # FloatList (packed floats)
                fl = decode_message(feature[2][0])
                vals = []
                for blob in fl.get(1, []):
                    if isinstance(blob, bytes):
                        import struct as _s
                        vals.extend(_s.unpack(f"<{len(blob) // 4}f", blob))
                    else:
                        vals.append(blob)
                feats[name] = vals
            elif 3 in feature:      # Int64List (packed varints)
                il = decode_message(feature[3][0])
                vals = []
                for blob in il.get(1, []):
                    if isinstance(blob, bytes):
                        i = 0
                        while i < len(blob):
                            v, i = _read_varint_signed(blob, i)
                            vals.append(v)
                    else:
                        vals.append(blob if blob < (1 << 63)
                                    else blob - (1 << 64))
                feats[name] = vals
        return feats

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, ...]]:
        rows = []
        for rec in self._records():
            feats = self._features(rec)
            label = float(feats.get("label", [0.0])[0])
            dense = []
            for i in range(N_DENSE):
                v = feats.get(f"I{i + 1}", [])
                dense.append(float(v[0]) if v else 0.0)
            cats = []
            for fidx in range(N_SPARSE):
                v = feats.get(f"C{fidx + 1}", [])
                if v and isinstance(v[0], bytes):
                    key = _hash_token(fidx, v[0].decode())
                    if not self.hash_mode:
                        key %= self.field_vocabs[fidx]
                elif v:
                    key = int(v[0])
                    if not self.hash_mode:
                        key %= self.field_vocabs[fidx]
                else:
                    key = 0 if not self.hash_mode else _hash_token(fidx, "")
                cats.append(key)
            rows.append((label, dense, cats))
            if len(rows) == self.batch_size:
                yield _emit_rows(rows)
                rows = []
        if rows and not self.drop_last:
            yield _emit_rows(rows)


def _read_varint_signed(blob: bytes, i: int):
    shift = 0
    v = 0
    while True:
        b = blob[i]
        i += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            if v >= (1 << 63):
                v -= 1 << 64
            return v, i
        shift += 7


def _emit_rows(rows) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    labels = torch.tensor([r[0] for r in rows], dtype=torch.float32)
    dense = torch.tensor([r[1] for r in rows], dtype=torch.float32)
    sparse = torch.tensor([r[2] for r in rows], dtype=torch.int64)
    return dense, sparse, labels


def write_tfrecord(path: str, examples: Iterable[Tuple[float, List[float],
                                                       List]]) -> int:
    """Write (label, dense[13], cats[26]) rows as a TFRecord of
    tf.train.Example — the converter half of the reference's
    criteo_preprocess step (and the fixture generator for tests). Cats may
    be ints (Int64List) or strings (BytesList tokens). Returns the record
    count."""
    import struct

    from .utils import tfproto as tp

    def feature_floats(vals):
        packed = b"".join(struct.pack("<f", float(v)) for v in vals)
        body = tp.tag(1, 2) + tp._varint(len(packed)) + packed
        return tp.f_msg(2, body)

    def feature_ints(vals):
        packed = b"".join(tp._varint(int(v) & ((1 << 64) - 1))
                          for v in vals)
        body = tp.tag(1, 2) + tp._varint(len(packed)) + packed
        return tp.f_msg(3, body)

    def feature_bytes(vals):
        return tp.f_msg(1, b"".join(tp.f_bytes(1, v) for v in vals))

    n = 0
    with open(path, "wb") as f:
        for label, dense, cats in examples:
            feats = b""
            feats += tp.f_map_entry(1, "label", feature_floats([label]))
            for i, v in enumerate(dense):
                feats += tp.f_map_entry(1, f"I{i + 1}",
                                        feature_floats([v]))
            for i, c in enumerate(cats):
                if isinstance(c, (bytes, str)):
                    cb = c.encode() if isinstance(c, str) else c
                    feats += tp.f_map_entry(1, f"C{i + 1}",
                                            feature_bytes([cb]))
                else:
                    feats += tp.f_map_entry(1, f"C{i + 1}",
                                            feature_ints([c]))
            example = tp.f_msg(1, feats)
            hdr = struct.pack("<Q", len(example))
            f.write(hdr)
            f.write(struct.pack("<I", tp.masked_crc32c(hdr)))
            f.write(example)
            f.write(struct.pack("<I", tp.masked_crc32c(example)))
            n += 1
    return n


class BackgroundLoader:
    """Producer-thread wrapper: parse/stage batches ``depth`` ahead of the
    train loop (the reference's dataset-thread prefetch half; pair with
    ``embed.pulling`` for the embedding-pull half). Optionally pins host
    memory so the H2D copies of the train loop are async-capable."""

    def __init__(self, source: Iterable, depth: int = 4,
                 pin_memory: bool = False):
        self.source = source
        self.depth = int(depth)
        self.pin = pin_memory and torch.cuda.is_available()

    def __iter__(self):
        q: "queue.Queue" = queue.Queue(maxsize=self.depth)
        _END = object()

        def produce():
            try:
                for item in self.source:
                    if self.pin:
                        item = tuple(t.pin_memory() for t in item)
                    q.put(item)
            finally:
                q.put(_END)

        t = threading.Thread(target=produce, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is _END:
                break
            yield item
        t.join()
