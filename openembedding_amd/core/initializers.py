"""Row initializers (reference openembedding/variable/EmbeddingInitializer.h).

Thin config wrappers over core.rng.init_rows; run lazily in the miss path of
pull/update (reference EmbeddingOptimizerVariable.h:255-262, :285-290)."""

from __future__ import annotations

from typing import Dict

import torch

from .rng import init_rows

__all__ = ["Initializer", "make_initializer", "INITIALIZERS"]


class Initializer:
    category = "base"
    defaults: Dict[str, float] = {}

    def __init__(self, **hyper):
        cfg = dict(self.defaults)
        for k, v in hyper.items():
            if k not in cfg:
                raise ValueError(
                    f"unknown hyperparameter {k!r} for initializer "
                    f"{self.category!r}; known: {sorted(cfg)}")
            cfg[k] = float(v)
        self.cfg = cfg

    def __call__(self, seed: int, keys: torch.Tensor, dim: int,
                 dtype: torch.dtype = torch.float32) -> torch.Tensor:
        return init_rows(self.category, self.cfg, seed, keys, dim, dtype)

    def dump_config(self):
        return dict(self.cfg)


class ConstantInitializer(Initializer):
    category = "constant"
    defaults = {"value": 0.0}


class UniformInitializer(Initializer):
    category = "uniform"
    defaults = {"minval": 0.0, "maxval": 1.0}


class NormalInitializer(Initializer):
    category = "normal"
    defaults = {"mean": 0.0, "stddev": 1.0, "truncated": 0.0}


INITIALIZERS = {cls.category: cls for cls in
                (ConstantInitializer, UniformInitializer, NormalInitializer)}


def make_initializer(category: str, **hyper) -> Initializer:
    if category not in INITIALIZERS:
        raise ValueError(f"unknown initializer {category!r}; "
                         f"known: {sorted(INITIALIZERS)}")
    return INITIALIZERS[category](**hyper)
