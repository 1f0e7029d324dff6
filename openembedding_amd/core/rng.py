"""Deterministic counter-based RNG for lazy row initialization.

The reference initializes missing rows with a host ``std::default_random_engine``
seeded from ``random_device`` (reference EmbeddingInitializer.h:40-49) — i.e.
non-reproducible. This framework improves on that: row init is a pure function
of ``(seed, key, column, attempt)`` via splitmix64, so

  - a key's initial row is identical no matter which GPU/shard materializes it,
    which makes checkpoints reload-stable across shard counts, and
  - the CPU torch path and the HIP kernel path produce bit-identical uniforms
    (both implement the same splitmix64; see ops/csrc/embops.hip).

All int64 arithmetic below relies on torch's wrapping (mod 2^64) semantics,
which matches uint64_t on device.
"""

from __future__ import annotations

import math

import torch

_C1 = -0x61C8864680B583EB  # 0x9E3779B97F4A7C15 as signed int64
_C2 = -0x40A7B892E31B1A47  # 0xBF58476D1CE4E5B9
_C3 = -0x6B2FB644ECCEEE15  # 0x94D049BB133111EB


def _lshr(z: torch.Tensor, k: int) -> torch.Tensor:
    """Logical (unsigned) right shift on int64 tensors."""
    return (z >> k) & ((1 << (64 - k)) - 1)


def splitmix64(x: torch.Tensor) -> torch.Tensor:
    """splitmix64 finalizer; x int64 tensor -> int64 tensor (uniform bits)."""
    z = x + _C1
    z = (z ^ _lshr(z, 30)) * _C2
    z = (z ^ _lshr(z, 27)) * _C3
    return z ^ _lshr(z, 31)


def uniform_bits(seed: int, key: torch.Tensor, col: torch.Tensor,
                 attempt: int = 0) -> torch.Tensor:
    """64 uniform bits per (key, col) pair. key/col broadcastable int64."""
    x = key * 0x100000 + col + (attempt << 40)
    return splitmix64(splitmix64(x) ^ seed)


def uniform01(seed: int, key: torch.Tensor, col: torch.Tensor,
              attempt: int = 0) -> torch.Tensor:
    """float32 uniform in [0,1): top 24 bits of splitmix64 scaled by 2^-24."""
    bits = _lshr(uniform_bits(seed, key, col, attempt), 40)
    return bits.to(torch.float32) * (1.0 / (1 << 24))


def init_rows(category: str, cfg: dict, seed: int, keys: torch.Tensor,
              dim: int, dtype: torch.dtype = torch.float32) -> torch.Tensor:
    """Materialize initial weight rows [n, dim] for the given keys.

    Matches the HIP miss-path initializer (ops/csrc/embops.hip init_rows_kernel)
    bit-for-bit for 'constant' and 'uniform'; 'normal' matches to float
    rounding (libm sin/cos/log differences).

    Semantics follow reference EmbeddingInitializer.h:
      constant: value (default 0)
      uniform:  [minval, maxval) (default [0,1))
      normal:   mean/stddev, with one-sided truncated rejection when
                truncated > 0.1 — resample while (w-mean)/stddev > truncated
                (reference :76-81 resamples on the upper side only).
    """
    n = keys.numel()
    device = keys.device
    if category == "constant":
        return torch.full((n, dim), float(cfg.get("value", 0.0)),
                          dtype=dtype, device=device)
    k = keys.view(-1, 1).expand(n, dim).to(torch.int64)
    c = torch.arange(dim, dtype=torch.int64, device=device).view(1, -1).expand(n, dim)
    if category == "uniform":
        lo = float(cfg.get("minval", 0.0))
        hi = float(cfg.get("maxval", 1.0))
        u = uniform01(seed, k, c)
        return (lo + u * (hi - lo)).to(dtype)
    if category == "normal":
        mean = float(cfg.get("mean", 0.0))
        stddev = float(cfg.get("stddev", 1.0))
        truncated = float(cfg.get("truncated", 0.0))
        out = _normal(seed, k, c, attempt=0)
        if truncated > 0.1:
            for attempt in range(1, 16):
                bad = out > truncated  # standardized sample above cut
                if not bool(bad.any()):
                    break
                out = torch.where(bad, _normal(seed, k, c, attempt=attempt), out)
        return (mean + stddev * out).to(dtype)
    raise ValueError(f"unknown initializer category {category!r}")


def _normal(seed: int, k: torch.Tensor, c: torch.Tensor, attempt: int) -> torch.Tensor:
    """Standard normal via Box-Muller from two splitmix64 uniforms."""
    u1 = uniform01(seed, k, c, attempt=2 * attempt)
    u2 = uniform01(seed, k, c, attempt=2 * attempt + 1)
    r = torch.sqrt(-2.0 * torch.log(1.0 - u1))  # 1-u1 in (0,1], avoids log(0)
    return r * torch.cos(2.0 * math.pi * u2)
