"""Streaming AUC/logloss vs sklearn exact references."""

import numpy as np
import pytest
import torch

from openembedding_amd.models.metrics import StreamingAUC, StreamingLogLoss


def test_auc_matches_sklearn():
    sk = pytest.importorskip("sklearn.metrics")
    g = torch.Generator().manual_seed(0)
    logits = torch.randn(20000, generator=g) * 2
    # correlated labels so AUC is well away from 0.5
    labels = (torch.sigmoid(logits + torch.randn(20000, generator=g))
              > 0.5).float()
    auc = StreamingAUC(bins=2000)
    for chunk in range(0, 20000, 3000):        # streamed in uneven chunks
        auc.update(logits[chunk:chunk + 3000], labels[chunk:chunk + 3000])
    ref = sk.roc_auc_score(labels.numpy(), torch.sigmoid(logits).numpy())
    assert abs(auc.compute() - ref) < 2e-3, (auc.compute(), ref)


def test_auc_degenerate_cases():
    auc = StreamingAUC(bins=100)
    assert 0.0 <= auc.compute() <= 1.0         # empty: defined
    auc.update(torch.tensor([2.0, -1.0]), torch.tensor([1.0, 1.0]))
    assert 0.0 <= auc.compute() <= 1.0         # single-class: no crash
    auc.reset()
    # perfectly separable
    auc.update(torch.tensor([5.0] * 50 + [-5.0] * 50),
               torch.tensor([1.0] * 50 + [0.0] * 50))
    assert auc.compute() > 0.99


def test_logloss_matches_torch():
    g = torch.Generator().manual_seed(1)
    logits = torch.randn(5000, generator=g)
    labels = (torch.rand(5000, generator=g) < 0.3).float()
    ll = StreamingLogLoss()
    for chunk in range(0, 5000, 700):
        ll.update(logits[chunk:chunk + 700], labels[chunk:chunk + 700])
    ref = torch.nn.functional.binary_cross_entropy_with_logits(
        logits, labels).item()
    assert abs(ll.compute() - ref) < 1e-6
