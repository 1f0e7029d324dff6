"""Optimizer numerics oracle (the reference's test/optimizer_test.py role).

Each sparse optimizer is checked against an INDEPENDENT scalar numpy
implementation of the reference formulas (EmbeddingOptimizer.h), over
multi-step gradient streams with duplicate keys, through the full
VariableShard pull/push/update path."""

import math

import numpy as np
import pytest
import torch

from openembedding_amd.core import VariableMeta, VariableShard, make_optimizer

DIM = 5


def scalar_reference(category, cfg, steps_grads, counts_per_step, w0):
    """Independent per-element reference. steps_grads: list of [dim] summed
    gradients; returns final weights [dim]."""
    w = np.array(w0, dtype=np.float32)
    dim = len(w)
    c = cfg
    if category == "adagrad":
        accum = np.full(dim, c["initial_accumulator_value"], np.float32)
    elif category in ("adadelta", "ftrl", "rmsprop"):
        if category == "ftrl":
            accum = np.full(dim, c["initial_accumulator_value"], np.float32)
            linear = np.zeros(dim, np.float32)
        else:
            accum = np.zeros(dim, np.float32)
            extra = np.zeros(dim, np.float32)
    elif category in ("adam", "adamax"):
        m = np.zeros(dim, np.float32)
        v = np.zeros(dim, np.float32)
        b1t = np.float32(1.0)
        b2t = np.float32(1.0)
    elif category == "sgd":
        moment = np.zeros(dim, np.float32)
    elif category == "test":
        s0 = np.float32(c.get("init", 0.0))

    for step, g in enumerate(steps_grads):
        g = np.asarray(g, np.float32)
        if category == "default":
            if c["learning_rate"] != 0:
                w = w - np.float32(c["learning_rate"]) * g
        elif category == "adagrad":
            accum = accum + g * g
            w = w - c["learning_rate"] * g / (np.sqrt(accum) + c["epsilon"])
        elif category == "adadelta":
            rho, eps, lr = c["rho"], c["epsilon"], c["learning_rate"]
            accum = accum * rho + g * g * (1 - rho)
            upd = g * np.sqrt(extra + eps) / np.sqrt(accum + eps)
            extra = extra * rho + upd * upd * (1 - rho)
            w = w - lr * upd
        elif category == "adam":
            lr, b1, b2, eps = (c["learning_rate"], c["beta_1"], c["beta_2"],
                               c["epsilon"])
            b1t *= np.float32(b1)
            b2t *= np.float32(b2)
            lr_t = lr * math.sqrt(1 - b2t) / (1 - b1t)
            m = m * b1 + g * (1 - b1)
            v = v * b2 + g * g * (1 - b2)
            w = w - np.float32(lr_t) * m / (np.sqrt(v) + eps)
        elif category == "adamax":
            lr, b1, b2, eps = (c["learning_rate"], c["beta_1"], c["beta_2"],
                               c["epsilon"])
            b1t *= np.float32(b1)
            lr_t = lr / (1 - b1t)
            m = m * b1 + g * (1 - b1)
            v = np.maximum(np.abs(g), v * b2)
            w = w - np.float32(lr_t) * m / (v + eps)
        elif category == "ftrl":
            lr = c["learning_rate"]
            l1 = c["l1_regularization_strength"]
            l2 = c["l2_regularization_strength"]
            l2s = c["l2_shrinkage_regularization_strength"]
            beta = c["beta"]
            adj_l2 = l2 + beta / lr / 2
            gg = g + 2 * l2s * w
            accum_new = accum + g * g
            sigma = (np.sqrt(accum_new) - np.sqrt(accum)) / lr
            linear = linear + gg - sigma * w
            accum = accum_new
            quad = np.sqrt(accum) / lr + 2 * adj_l2
            l1a = np.clip(linear, -l1, l1)
            w = (l1a - linear) / quad
        elif category == "rmsprop":
            lr, rho, mom, eps = (c["learning_rate"], c["rho"], c["momentum"],
                                 c["epsilon"])
            accum = accum * rho + g * g * (1 - rho)
            extra = extra * mom + lr * g / np.sqrt(accum + eps)
            w = w - extra
        elif category == "sgd":
            lr, mom = c["learning_rate"], c["momentum"]
            moment = moment * mom + lr * g
            if c["nesterov"]:
                w = w - (moment * mom + lr * g)
            else:
                w = w - moment
        elif category == "test":
            s0 = np.float32(c["flip"]) - s0
            w = w + c["learning_rate"] * g / counts_per_step[step] + s0
    return w


CONFIGS = [
    ("default", {"learning_rate": 0.05}),
    ("adagrad", {}),
    ("adagrad", {"learning_rate": 0.01, "initial_accumulator_value": 0.5}),
    ("adadelta", {}),
    ("adadelta", {"rho": 0.8, "learning_rate": 0.1}),
    ("adam", {}),
    ("adam", {"beta_1": 0.85, "beta_2": 0.99, "learning_rate": 0.01}),
    ("adamax", {}),
    ("ftrl", {}),
    ("ftrl", {"l1_regularization_strength": 0.01,
              "l2_regularization_strength": 0.02,
              "l2_shrinkage_regularization_strength": 0.01, "beta": 0.5}),
    ("rmsprop", {}),
    ("rmsprop", {"momentum": 0.5}),
    ("sgd", {}),
    ("sgd", {"momentum": 0.9}),
    ("sgd", {"momentum": 0.9, "nesterov": True}),
    ("test", {}),
]


@pytest.mark.parametrize("category,cfg", CONFIGS)
@pytest.mark.parametrize("steps", [1, 3, 10])
def test_optimizer_vs_scalar_reference(category, cfg, steps):
    torch.manual_seed(42)
    meta = VariableMeta(variable_id=0, embedding_dim=DIM, vocabulary_size=100)
    shard = VariableShard(meta, 0, 1, device="cpu", seed=7)
    shard.set_initializer("uniform", minval=-0.5, maxval=0.5)
    shard.set_optimizer(category, **cfg)
    full_cfg = shard.optimizer.cfg

    key = torch.tensor([17], dtype=torch.int64)
    w0 = shard.pull(key)[0].numpy().copy()

    steps_grads = []
    counts_per_step = []
    for s in range(steps):
        # duplicate key 3x in the batch: grads should be SUMMED, count = 3
        g = torch.randn(3, DIM)
        keys3 = key.repeat(3)
        shard.pull(keys3.unique())
        shard.push(key, g.sum(0, keepdim=True),
                   torch.tensor([3], dtype=torch.int64))
        shard.update_weights()
        steps_grads.append(g.sum(0).numpy())
        counts_per_step.append(3)

    expected = scalar_reference(category, full_cfg, steps_grads,
                                counts_per_step, w0)
    got = shard.pull_readonly(key)[0].numpy()
    np.testing.assert_allclose(got, expected, rtol=2e-5, atol=2e-6)


def test_unknown_hyperparameter_rejected():
    with pytest.raises(ValueError):
        make_optimizer("adam", bogus=1.0)
    with pytest.raises(ValueError):
        make_optimizer("nope")


def test_state_dims():
    d = 8
    expect = {"default": 0, "adadelta": 16, "adagrad": 8, "adam": 18,
              "adamax": 17, "ftrl": 16, "rmsprop": 16, "sgd": 8, "test": 2}
    for cat, sd in expect.items():
        assert make_optimizer(cat).state_dim(d) == sd, cat
