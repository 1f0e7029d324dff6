"""Sparse-embedding optimizers — the numerics contract of the framework.

Formula-for-formula the semantics of the reference's server-side optimizers
(reference openembedding/variable/EmbeddingOptimizer.h:49-390), vectorized as
torch ops over a batch of unique rows. This module is the CPU execution path
AND the oracle the fused HIP kernels (ops/csrc/optimizers.hip) are tested
against.

Contract (reference EmbeddingOptimizerVariable.h:273-297 + MpscGradientReducer.h):
  - gradients arriving for one batch are SUMMED per unique key (no averaging),
    with an occurrence ``count`` kept per key;
  - the optimizer steps ONCE per unique key per committed batch;
  - only the deterministic ``test`` optimizer divides by count;
  - the row layout is ``weights[dim] || state[state_dim]`` contiguous fp32/fp64;
  - state is initialized by ``train_init`` when the row first joins the table.

State layouts (offsets into the state part of the row), per reference
``state_dim`` definitions:
  default : []                                  (EmbeddingOptimizer.h:49-72)
  adadelta: [accum(dim), accum_update(dim)]     (:76-113)
  adagrad : [accum(dim)]                        (:117-144)
  adam    : [m(dim), v(dim), beta_1_t, beta_2_t](:148-187)
  adamax  : [m(dim), v(dim), beta_1_t]          (:191-226)
  ftrl    : [accum(dim), linear(dim)]           (:230-293)
  rmsprop : [accum(dim), moment(dim)]           (:297-328)
  sgd     : [moment(dim)]                       (:332-363)
  test    : [flip_state, pad]                   (:367-390)
"""

from __future__ import annotations

import math
from typing import Dict

import torch

__all__ = ["OPTIMIZERS", "SparseOptimizer", "make_optimizer"]


class SparseOptimizer:
    """Base: a named sparse optimizer with a typed hyper-parameter dict."""

    category = "base"
    defaults: Dict[str, float] = {}

    def __init__(self, **hyper):
        cfg = dict(self.defaults)
        for k, v in hyper.items():
            if k not in cfg:
                raise ValueError(
                    f"unknown hyperparameter {k!r} for optimizer {self.category!r}; "
                    f"known: {sorted(cfg)}"
                )
            cfg[k] = type(cfg[k])(v) if not isinstance(cfg[k], bool) else bool(v)
        self.cfg = cfg

    def state_dim(self, dim: int) -> int:
        raise NotImplementedError

    def train_init(self, state: torch.Tensor, dim: int) -> None:
        """Initialize optimizer state in-place. state: [n, state_dim]."""
        state.zero_()

    def update(self, w: torch.Tensor, s: torch.Tensor, counts: torch.Tensor,
               g: torch.Tensor) -> None:
        """Apply one optimizer step in-place.

        w: [n, dim] weights; s: [n, state_dim] state; counts: [n] int64
        occurrence counts; g: [n, dim] summed gradients.
        """
        raise NotImplementedError

    # ---- serialization of hyper-params (checkpoint config parity) ----
    def dump_config(self) -> Dict[str, float]:
        return dict(self.cfg)


class DefaultOptimizer(SparseOptimizer):
    """Plain SGD step, stateless (reference 'default', EmbeddingOptimizer.h:49-72)."""

    category = "default"
    defaults = {"learning_rate": 0.0}

    def state_dim(self, dim):
        return 0

    def update(self, w, s, counts, g):
        lr = self.cfg["learning_rate"]
        if lr != 0:
            w.sub_(g, alpha=lr)


class AdadeltaOptimizer(SparseOptimizer):
    category = "adadelta"
    defaults = {"learning_rate": 0.001, "rho": 0.95, "epsilon": 1e-7}

    def state_dim(self, dim):
        return 2 * dim

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        lr, rho, eps = (self.cfg[k] for k in ("learning_rate", "rho", "epsilon"))
        accum = s[:, :dim]
        accum_update = s[:, dim:2 * dim]
        accum.mul_(rho).add_(g * g, alpha=1 - rho)
        update = g * torch.sqrt(accum_update + eps) / torch.sqrt(accum + eps)
        accum_update.mul_(rho).add_(update * update, alpha=1 - rho)
        w.sub_(update, alpha=lr)


class AdagradOptimizer(SparseOptimizer):
    category = "adagrad"
    defaults = {"learning_rate": 0.001, "initial_accumulator_value": 0.1,
                "epsilon": 1e-7}

    def state_dim(self, dim):
        return dim

    def train_init(self, state, dim):
        state.fill_(self.cfg["initial_accumulator_value"])

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        lr, eps = self.cfg["learning_rate"], self.cfg["epsilon"]
        accum = s[:, :dim]
        accum.add_(g * g)
        w.sub_(lr * g / (torch.sqrt(accum) + eps))


class AdamOptimizer(SparseOptimizer):
    """Adam with PER-ROW beta^t power state (reference EmbeddingOptimizer.h:148-187):
    each row's bias correction advances once per batch the row was touched in,
    not per global step."""

    category = "adam"
    defaults = {"learning_rate": 0.001, "beta_1": 0.9, "beta_2": 0.999,
                "epsilon": 1e-7}

    def state_dim(self, dim):
        return 2 * dim + 2

    def train_init(self, state, dim):
        state.zero_()
        state[:, 2 * dim] = 1.0
        state[:, 2 * dim + 1] = 1.0

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        lr, b1, b2, eps = (self.cfg[k] for k in
                           ("learning_rate", "beta_1", "beta_2", "epsilon"))
        m = s[:, :dim]
        v = s[:, dim:2 * dim]
        b1t = s[:, 2 * dim]
        b2t = s[:, 2 * dim + 1]
        b1t.mul_(b1)
        b2t.mul_(b2)
        lr_t = lr * torch.sqrt(1 - b2t) / (1 - b1t)  # [n]
        m.mul_(b1).add_(g, alpha=1 - b1)
        v.mul_(b2).add_(g * g, alpha=1 - b2)
        w.sub_(lr_t.unsqueeze(1) * m / (torch.sqrt(v) + eps))


class AdamaxOptimizer(SparseOptimizer):
    category = "adamax"
    defaults = {"learning_rate": 0.001, "beta_1": 0.9, "beta_2": 0.999,
                "epsilon": 1e-7}

    def state_dim(self, dim):
        return 2 * dim + 1

    def train_init(self, state, dim):
        state.zero_()
        state[:, 2 * dim] = 1.0

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        lr, b1, b2, eps = (self.cfg[k] for k in
                           ("learning_rate", "beta_1", "beta_2", "epsilon"))
        m = s[:, :dim]
        v = s[:, dim:2 * dim]
        b1t = s[:, 2 * dim]
        b1t.mul_(b1)
        lr_t = lr / (1 - b1t)  # [n]
        m.mul_(b1).add_(g, alpha=1 - b1)
        torch.maximum(g.abs(), v * b2, out=v)
        w.sub_(lr_t.unsqueeze(1) * m / (v + eps))


class FtrlOptimizer(SparseOptimizer):
    category = "ftrl"
    defaults = {"learning_rate": 0.001, "initial_accumulator_value": 0.1,
                "l1_regularization_strength": 0.0,
                "l2_regularization_strength": 0.0,
                "l2_shrinkage_regularization_strength": 0.0,
                "learning_rate_power": -0.5, "beta": 0.0}

    def state_dim(self, dim):
        return 2 * dim

    def train_init(self, state, dim):
        state[:, :dim] = self.cfg["initial_accumulator_value"]
        state[:, dim:] = 0.0

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        c = self.cfg
        lr = c["learning_rate"]
        accum = s[:, :dim]
        linear = s[:, dim:2 * dim]
        adj_l2 = c["l2_regularization_strength"] + c["beta"] / lr / 2
        gg = g + 2 * c["l2_shrinkage_regularization_strength"] * w
        accum_new = accum + g * g  # reference sums grad*grad, not gg*gg
        p = -c["learning_rate_power"]
        if c["learning_rate_power"] == -0.5:
            sigma = (torch.sqrt(accum_new) - torch.sqrt(accum)) / lr
            linear.add_(gg - sigma * w)
            accum.copy_(accum_new)
            quadratic = torch.sqrt(accum) / lr + 2 * adj_l2
        else:
            sigma = (accum_new.pow(p) - accum.pow(p)) / lr
            linear.add_(gg - sigma * w)
            accum.copy_(accum_new)
            quadratic = accum.pow(p) / lr + 2 * adj_l2
        l1 = c["l1_regularization_strength"]
        l1_adjust = linear.clamp(min=-l1, max=l1)
        w.copy_((l1_adjust - linear) / quadratic)


class RMSpropOptimizer(SparseOptimizer):
    category = "rmsprop"
    defaults = {"learning_rate": 0.001, "rho": 0.9, "momentum": 0.0,
                "epsilon": 1e-7}

    def state_dim(self, dim):
        return 2 * dim

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        lr, rho, mom, eps = (self.cfg[k] for k in
                             ("learning_rate", "rho", "momentum", "epsilon"))
        accum = s[:, :dim]
        moment = s[:, dim:2 * dim]
        accum.mul_(rho).add_(g * g, alpha=1 - rho)
        moment.mul_(mom).add_(lr * g / torch.sqrt(accum + eps))
        w.sub_(moment)


class SGDOptimizer(SparseOptimizer):
    category = "sgd"
    defaults = {"learning_rate": 0.01, "momentum": 0.0, "nesterov": False}

    def state_dim(self, dim):
        return dim

    def update(self, w, s, counts, g):
        dim = w.shape[1]
        lr, mom = self.cfg["learning_rate"], self.cfg["momentum"]
        moment = s[:, :dim]
        moment.mul_(mom).add_(g, alpha=lr)
        if self.cfg["nesterov"]:
            w.sub_(moment * mom + lr * g)
        else:
            w.sub_(moment)


class TestOptimizer(SparseOptimizer):
    """Deterministic optimizer for unit tests (reference
    EmbeddingOptimizer.h:367-390): flip-flop state so missed/double updates
    are detected; divides the summed gradient by the occurrence count."""

    category = "test"
    defaults = {"learning_rate": 0.1, "flip": 10000.0, "init": 0.0}

    def state_dim(self, dim):
        return 2

    def train_init(self, state, dim):
        state.zero_()
        state[:, 0] = self.cfg["init"]

    def update(self, w, s, counts, g):
        lr, flip = self.cfg["learning_rate"], self.cfg["flip"]
        s[:, 0] = flip - s[:, 0]
        w.add_(lr * g / counts.unsqueeze(1).to(g.dtype) + s[:, 0].unsqueeze(1))


OPTIMIZERS = {
    cls.category: cls
    for cls in (DefaultOptimizer, AdadeltaOptimizer, AdagradOptimizer,
                AdamOptimizer, AdamaxOptimizer, FtrlOptimizer,
                RMSpropOptimizer, SGDOptimizer, TestOptimizer)
}


def make_optimizer(category: str, **hyper) -> SparseOptimizer:
    if category not in OPTIMIZERS:
        raise ValueError(f"unknown sparse optimizer {category!r}; "
                         f"known: {sorted(OPTIMIZERS)}")
    return OPTIMIZERS[category](**hyper)
