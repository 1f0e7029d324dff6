"""Streaming evaluation metrics for CTR training.

The reference's benchmark tracked AUC + binary cross-entropy through Keras
metrics (test/benchmark/criteo_deepctr.py model.compile(metrics=[AUC()])).
These are the torch-side equivalents, streaming (constant memory per
batch) and collective-aware: ``sync()`` merges the accumulated state across
ranks so every rank reports the global metric.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def _reduce_tensor(t: torch.Tensor) -> torch.Tensor:
    """all_reduce(SUM) on a device the process-group backend accepts: the
    nccl/RCCL backend rejects CPU tensors, so accumulator state kept on CPU
    is staged through the current GPU for the reduction."""
    if not (dist.is_available() and dist.is_initialized()):
        return t
    if dist.get_backend() == "nccl" and t.device.type != "cuda":
        d = t.to("cuda")
        dist.all_reduce(d, op=dist.ReduceOp.SUM)
        return d.to(t.device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


class StreamingAUC:
    """Histogram-bucketed ROC-AUC over logits (constant memory).

    Scores are bucketed by sigmoid(logit) into ``bins`` equal-width buckets;
    AUC is the trapezoidal area under the ROC curve of the bucket
    boundaries — the same estimator Keras' `tf.keras.metrics.AUC`
    (num_thresholds=bins) uses, so parity with the reference's reported
    numbers holds at equal bin counts.
    """

    def __init__(self, bins: int = 1000, device: str = "cpu"):
        self.bins = int(bins)
        self.pos = torch.zeros(self.bins, dtype=torch.float64, device=device)
        self.neg = torch.zeros(self.bins, dtype=torch.float64, device=device)

    @torch.no_grad()
    def update(self, logits: torch.Tensor, labels: torch.Tensor) -> None:
        p = torch.sigmoid(logits.detach().float()).clamp(0, 1 - 1e-9)
        b = (p * self.bins).long()
        y = labels.reshape(-1).float()
        self.pos += torch.bincount(b, weights=y, minlength=self.bins
                                   ).to(self.pos)
        self.neg += torch.bincount(b, weights=1.0 - y, minlength=self.bins
                                   ).to(self.neg)

    def sync(self) -> "StreamingAUC":
        if dist.is_available() and dist.is_initialized():
            self.pos = _reduce_tensor(self.pos)
            self.neg = _reduce_tensor(self.neg)
        return self

    def compute(self) -> float:
        # sweep thresholds from high to low: cumulative TP/FP from the top
        pos = torch.flip(self.pos, [0]).cumsum(0)
        neg = torch.flip(self.neg, [0]).cumsum(0)
        P = float(pos[-1]) or 1.0
        N = float(neg[-1]) or 1.0
        tpr = torch.cat([torch.zeros(1, dtype=torch.float64,
                                     device=pos.device), pos / P])
        fpr = torch.cat([torch.zeros(1, dtype=torch.float64,
                                     device=neg.device), neg / N])
        return float(torch.trapz(tpr, fpr))

    def reset(self) -> None:
        self.pos.zero_()
        self.neg.zero_()


class StreamingLogLoss:
    """Mean binary cross-entropy over logits, streaming."""

    def __init__(self):
        self.total = 0.0
        self.n = 0

    @torch.no_grad()
    def update(self, logits: torch.Tensor, labels: torch.Tensor) -> None:
        z = logits.detach().float().reshape(-1)
        y = labels.reshape(-1).float()
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            z, y, reduction="sum")
        self.total += float(loss)
        self.n += z.numel()

    def sync(self) -> "StreamingLogLoss":
        if dist.is_available() and dist.is_initialized():
            t = torch.tensor([self.total, float(self.n)], dtype=torch.float64)
            t = _reduce_tensor(t)
            self.total, self.n = float(t[0]), int(t[1])
        return self

    def compute(self) -> float:
        return self.total / max(self.n, 1)
