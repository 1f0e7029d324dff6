"""Criteo-shaped synthetic data (13 dense + 26 categorical features).

The reference benchmarks on Criteo Kaggle (documents/en/benchmark.md:7-17);
there is no network here, so models train on synthetic batches with the
well-known Criteo-Kaggle per-field cardinalities (~33.8M ids total) and
random labels. Field ids are uniform within each field — conservative for
the engine (less dedup than the real power-law distribution)."""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

# Criteo Kaggle (dac) categorical cardinalities, C1..C26.
CRITEO_FIELD_VOCABS: List[int] = [
    1460, 583, 10131227, 2202608, 305, 24, 12517, 633, 3, 93145, 5683,
    8351593, 3194, 27, 14992, 5461306, 10, 5652, 2173, 4, 7046547, 18, 15,
    286181, 105, 142572,
]
N_DENSE = 13
N_SPARSE = len(CRITEO_FIELD_VOCABS)


def synthetic_batch(batch_size: int, device: str = "cpu",
                    field_vocabs: Optional[List[int]] = None,
                    generator: Optional[torch.Generator] = None
                    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """-> (dense [B,13] float32, sparse [B,26] int64 per-field ids,
    labels [B] float32)."""
    fv = field_vocabs or CRITEO_FIELD_VOCABS
    dense = torch.rand(batch_size, N_DENSE, device=device, generator=generator)
    cols = [torch.randint(0, v, (batch_size,), device=device, dtype=torch.int64,
                          generator=generator) for v in fv]
    sparse = torch.stack(cols, dim=1)
    labels = (torch.rand(batch_size, device=device, generator=generator)
              < 0.25).float()
    return dense, sparse, labels
