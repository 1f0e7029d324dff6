"""Prefetch pipeline (reference pulling() + PrefetchPullWeights,
exb.py:645-691, exb_ops.cpp:109-205): ahead-of-time pulls must consume at
forward time and training must be numerically identical to the
non-prefetched run (the reference's ordering guarantee via the batch-id
pending queue; here via issue-after-commit stream order)."""

import torch

import openembedding_amd.torch as embed
from openembedding_amd.models import DeepFM, synthetic_batch


def _train(model, opt, batches, use_pulling):
    lossf = torch.nn.BCEWithLogitsLoss()
    losses = []
    it = (embed.pulling(batches, model) if use_pulling else iter(batches))
    for dense, sparse, labels in it:
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def _make(seed=0):
    torch.manual_seed(seed)
    model = DeepFM(dim=4)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01))
    return model, opt


def _batches(n=6, bs=64):
    gen = torch.Generator().manual_seed(42)
    return [synthetic_batch(bs, generator=gen) for _ in range(n)]


def test_pulling_matches_plain_training():
    batches = _batches()
    m1, o1 = _make()
    plain = _train(m1, o1, batches, use_pulling=False)

    import openembedding_amd.context as cm
    import openembedding_amd.torch as api
    cm._context.finalize()
    cm._context = None
    api._tracked.clear()

    m2, o2 = _make()
    pref = _train(m2, o2, batches, use_pulling=True)
    assert plain == pref  # bitwise-identical losses


def test_prefetch_consumed_fifo():
    model, opt = _make(seed=1)
    emb = model.embedding
    b = _batches(3)
    for dense, sparse, labels in b:
        emb.prefetch(sparse)
    assert len(emb.variable._prefetched) == 3
    out = model(b[0][0], b[0][1])
    assert len(emb.variable._prefetched) == 2
    # skipping batch 1: consuming batch 2 drops the stale entry
    model(b[2][0], b[2][1])
    assert len(emb.variable._prefetched) == 0
    assert out.shape == (64, 1) or out.dim() == 1


def test_unmatched_prefetch_falls_through():
    model, opt = _make(seed=2)
    emb = model.embedding
    dense, sparse, labels = synthetic_batch(32)
    other = synthetic_batch(32)[1]
    emb.prefetch(other)
    out = emb(sparse)            # different tensor: normal pull path
    assert len(emb.variable._prefetched) == 1
    ref = emb.variable.sparse_read(sparse + emb.field_offsets)
    assert torch.allclose(out.detach(), ref)


def test_pulling_without_model_noop():
    batches = _batches(2)
    got = list(embed.pulling(batches))
    assert len(got) == 2
