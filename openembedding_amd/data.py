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
