"""End-to-end CTR models on CPU: forward/backward/step must run and learn."""

import pytest
import torch

import openembedding_amd.torch as embed
from openembedding_amd.models import MODELS, synthetic_batch


FIELD_VOCABS = [50, 3, 1000, 40, 7] + [100] * 21  # small criteo-shaped


@pytest.mark.parametrize("name", ["lr", "wdl", "deepfm", "xdeepfm"])
def test_model_trains(name):
    torch.manual_seed(0)
    kw = {} if name == "lr" else {"dim": 4}
    model = MODELS[name](field_vocabs=FIELD_VOCABS, **kw)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad([p for p in model.parameters() if p.numel()],
                            lr=0.05))
    lossf = torch.nn.BCEWithLogitsLoss()
    first = last = None
    g = torch.Generator().manual_seed(1)
    dense, sparse, labels = synthetic_batch(256, field_vocabs=FIELD_VOCABS,
                                            generator=g)
    # fixed batch -> loss must drop if grads flow through embeddings
    for step in range(12):
        opt.zero_grad()
        out = model(dense, sparse)
        loss = lossf(out, labels)
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
        last = loss.item()
    assert last < first, (first, last)


def test_embedding_gradient_flow():
    torch.manual_seed(0)
    e = embed.Embedding(100, 8)
    e.variable.set_optimizer("sgd", learning_rate=0.5)
    idx = torch.tensor([[3, 3, 4]], dtype=torch.int64)
    out = e(idx)
    assert out.shape == (1, 3, 8)
    before = e.variable.sparse_read(torch.tensor([3, 4]))
    out.sum().backward()
    embed.get_context().update_all_weights()
    after = e.variable.sparse_read(torch.tensor([3, 4]))
    # key 3 appears twice: summed grad 2*ones -> w -= 0.5*2; key 4 -> w -= 0.5
    torch.testing.assert_close(after[0], before[0] - 1.0)
    torch.testing.assert_close(after[1], before[1] - 0.5)


def test_hash_mode_embedding():
    e = embed.Embedding(-1, 4)
    e.variable.set_optimizer("sgd", learning_rate=0.1)
    idx = torch.tensor([123456789012345, 5], dtype=torch.int64)
    out = e(idx)
    out.sum().backward()
    embed.get_context().update_all_weights()
    assert e.variable.sharded.shard.num_rows == 2


def test_distributed_model_surgery():
    m = torch.nn.Sequential()
    m.add_module("big", torch.nn.Embedding(1000, 8))
    m.add_module("small", torch.nn.Embedding(10, 8))
    embed.distributed_model(m, sparse_as_dense_size=64)
    assert isinstance(m.big, embed.Embedding)
    assert isinstance(m.small, torch.nn.Embedding)


def test_sparse_as_dense_cache_path():
    e = embed.Embedding(10, 4, sparse_as_dense=True)
    out = e(torch.tensor([1, 2]))
    assert out.requires_grad and out.shape == (2, 4)


def test_optimizer_class_surface():
    """Reference exb exports optimizer classes (exb.py:446-488):
    embed.Adagrad(...) pre-wraps distributed_optimizer."""
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch

    torch.manual_seed(0)
    m = DeepFM(dim=4)
    opt = embed.Adagrad(m.parameters(), lr=0.01)
    assert isinstance(opt, embed.DistributedOptimizer)
    dense, sparse, labels = synthetic_batch(32)
    loss = torch.nn.functional.binary_cross_entropy_with_logits(
        m(dense, sparse), labels)
    loss.backward()
    opt.step()

    m2 = DeepFM(dim=4)
    o2 = embed.Ftrl(m2.parameters(), learning_rate=0.05,
                    l1_regularization_strength=0.01)
    loss = torch.nn.functional.binary_cross_entropy_with_logits(
        m2(*synthetic_batch(32)[:2]), synthetic_batch(32)[2])
    loss.backward()
    o2.step()
    assert m2.embedding.variable.sharded.shard.optimizer.category == "ftrl"


@pytest.mark.parametrize("batch", [1, 3, 10, 50])
def test_tiny_and_odd_batches(batch):
    # the reference's one-batch edge cases (build.sh test: batch 100/50/10);
    # odd shapes must survive the whole dedup/all_to_all/gather/reduce stack
    torch.manual_seed(1)
    model = MODELS["deepfm"](field_vocabs=FIELD_VOCABS, dim=4)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad([p for p in model.parameters() if p.numel()],
                            lr=0.05))
    for _ in range(3):
        dense, sparse, labels = synthetic_batch(batch)
        sparse = sparse % torch.tensor(FIELD_VOCABS)
        opt.zero_grad()
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            model(dense, sparse), labels)
        loss.backward()
        opt.step()
    assert torch.isfinite(loss)


def test_repeated_key_batch():
    # an entire batch hitting ONE key per field: counts path + once-per-key
    # optimizer application under maximal duplication
    torch.manual_seed(2)
    model = MODELS["deepfm"](field_vocabs=FIELD_VOCABS, dim=4)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad([p for p in model.parameters() if p.numel()],
                            lr=0.05))
    dense = torch.rand(64, 13)
    sparse = torch.zeros(64, 26, dtype=torch.int64)    # all-duplicate keys
    labels = torch.ones(64)
    for _ in range(2):
        opt.zero_grad()
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            model(dense, sparse), labels)
        loss.backward()
        opt.step()
    assert torch.isfinite(loss)
