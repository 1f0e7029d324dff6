"""Backend dispatch for the engine's hot ops.

Each op has a pure-torch implementation (CPU oracle; exact reference
semantics) and routes to the HIP extension on ROCm devices. The HIP kernels
are tested against the torch versions in tests/test_gpu_numerics.py.
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import require_hip


def _use_hip(t: torch.Tensor) -> bool:
    return t.is_cuda


# --------------------------------------------------------------------- unique

def unique_inverse(keys: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Batch-local dedup: keys int64 [n] -> (unique [u], inverse [n]).

    GPU: hash-based (no sort) CDNA4 kernel — the reference's client-side
    dedup (EmbeddingPullOperator.cpp:67-79) moved on-device. Unique order is
    unspecified (GPU: first-probe order; CPU: sorted)."""
    if keys.numel() == 0:
        return keys.clone(), torch.empty(0, dtype=torch.int64, device=keys.device)
    if _use_hip(keys):
        ext = require_hip()
        if ext is not None:
            return ext.unique_inverse(keys)
    return torch.unique(keys, return_inverse=True)


# ------------------------------------------------------------- reduce-by-key

def reduce_by_inverse(inverse: torch.Tensor, grads: torch.Tensor, u: int
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Sum gradients per unique id + occurrence counts (the reference's
    client pre-aggregation, EmbeddingPushOperator.cpp:39-58).

    inverse int64 [n], grads [n, dim] -> (ugrads [u, dim], counts int64 [u]).
    """
    if _use_hip(grads) and grads.dtype == torch.float32:
        ext = require_hip()   # f64 variables use the torch path below
        if ext is not None:
            return ext.reduce_by_inverse(inverse, grads, u)
    ugrads = torch.zeros((u, grads.shape[1]), dtype=grads.dtype,
                         device=grads.device)
    ugrads.index_add_(0, inverse, grads)
    counts = torch.bincount(inverse, minlength=u).to(torch.int64)
    return ugrads, counts


# ----------------------------------------------------------------- fused loss

class _FusedBCEFn(torch.autograd.Function):
    """BCEWithLogitsLoss(mean) in one kernel per direction at bench-size
    batches — a single-block forward with a plain store (no pre-fill, no
    atomics) up to n=64k, the fill+atomic grid form above that
    (torch spends ~5 launches per
    step on it inside the captured train graph: log_sigmoid + mean reduce +
    grad fill + the sigmoid-sub-scale chain). Same stable formulation, so
    it matches torch to fp32 atomic-order noise."""

    @staticmethod
    def forward(ctx, logits, labels, ext):
        logits = logits.contiguous()
        labels = labels.contiguous()
        ctx.save_for_backward(logits, labels)
        return ext.bce_fwd(logits, labels)

    @staticmethod
    def backward(ctx, grad_out):
        logits, labels = ctx.saved_tensors
        ext = require_hip()
        g = ext.bce_bwd(logits, labels, grad_out)
        return g, None, None


def bce_with_logits(logits: torch.Tensor, labels: torch.Tensor
                    ) -> torch.Tensor:
    """Mean binary-cross-entropy-with-logits; fused HIP kernels on GPU,
    torch elsewhere. Differentiable w.r.t. logits."""
    if _use_hip(logits) and logits.dtype == torch.float32:
        ext = require_hip()
        if ext is not None:
            return _FusedBCEFn.apply(logits, labels.float(), ext)
    return torch.nn.functional.binary_cross_entropy_with_logits(
        logits, labels.to(logits.dtype))
