"""hipGraph capture/replay correctness: a captured DeepFM train step
replayed with FRESH data must produce the same weights as eager execution.
Regression guard for the class of bugs where work issued inside capture is
not replayed (e.g. hipMemsetAsync scratch clears — see embops.hip fills)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _build():
    import openembedding_amd.context as cm
    import openembedding_amd.torch as api
    if cm._context is not None:
        cm._context.finalize()
        cm._context = None
    api._tracked.clear()
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM

    torch.manual_seed(0)
    model = DeepFM(dim=9).to(DEV)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01),
        flatten_dense=True)
    return model, opt


def _batches(n):
    from openembedding_amd.models import synthetic_batch
    gen = torch.Generator().manual_seed(77)
    return [tuple(t.to(DEV) for t in synthetic_batch(1024, generator=gen))
            for _ in range(n)]


def test_graph_replay_matches_eager():
    lossf = torch.nn.BCEWithLogitsLoss()
    batches = _batches(6)

    # eager reference
    model, opt = _build()

    def step(m, o, dense, sparse, labels):
        o.zero_grad(set_to_none=False)
        loss = lossf(m(dense, sparse), labels)
        loss.backward()
        o.step()

    # the graph path warms up with 6 steps on batch0 (3 eager + 3 on a side
    # stream) before capture; mirror the exact sequence here eagerly
    for _ in range(6):
        step(model, opt, *batches[0])
    for b in batches:
        step(model, opt, *b)
    probe = torch.arange(0, 2000, 7, dtype=torch.int64, device=DEV)
    ref_rows = model.embedding.variable.sparse_read(probe).clone()
    ref_dense = torch.cat([p.detach().reshape(-1).float()
                           for p in model.dnn.parameters()]).clone()

    # graph-captured run over the same batches
    model2, opt2 = _build()
    static = tuple(t.clone() for t in batches[0])
    for _ in range(3):
        step(model2, opt2, *static)
    torch.cuda.synchronize()
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            step(model2, opt2, *static)
    torch.cuda.current_stream().wait_stream(side)
    g = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g):
            step(model2, opt2, *static)
    except RuntimeError:
        pytest.skip("capture unavailable for this configuration")
    # capture records without executing; replay every batch once
    for b in batches:
        static[0].copy_(b[0])
        static[1].copy_(b[1])
        static[2].copy_(b[2])
        g.replay()
    torch.cuda.synchronize()
    got_rows = model2.embedding.variable.sparse_read(probe)
    got_dense = torch.cat([p.detach().reshape(-1).float()
                           for p in model2.dnn.parameters()])
    # float atomics (head backward dw/db, reduce-by-key) make two separate
    # runs differ at rounding level, and 12 Adagrad steps amplify a 1-ulp
    # grad flip into ~1e-4..1e-3-scale weight deltas (observed flaking at
    # 1e-4 and, rarely, at 1e-3 depending on which tests ran before); a
    # replay bug (e.g. a non-replayed scratch reset) shows up orders of
    # magnitude above this
    assert torch.allclose(got_rows, ref_rows, atol=5e-3, rtol=5e-3), \
        float((got_rows - ref_rows).abs().max())
    assert torch.allclose(got_dense, ref_dense, atol=5e-3, rtol=5e-3), \
        float((got_dense - ref_dense).abs().max())


def test_graph_replay_hash_mode_with_reservation():
    """CombinedEmbedding hash mode auto-reserves its key space, making the
    insert path capturable; replayed training must equal eager."""
    import openembedding_amd.context as cm
    import openembedding_amd.torch as api
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch

    lossf = torch.nn.BCEWithLogitsLoss()
    gen = torch.Generator().manual_seed(5)
    batches = [tuple(t.to(DEV) for t in synthetic_batch(512, generator=gen))
               for _ in range(5)]

    def build():
        if cm._context is not None:
            cm._context.finalize()
            cm._context = None
        api._tracked.clear()
        torch.manual_seed(0)
        m = DeepFM(dim=4, hash_mode=True).to(DEV)
        o = embed.distributed_optimizer(
            torch.optim.Adagrad(m.parameters(), lr=0.01), flatten_dense=True)
        return m, o

    def step(m, o, dense, sparse, labels):
        o.zero_grad(set_to_none=False)
        loss = lossf(m(dense, sparse), labels)
        loss.backward()
        o.step()

    m1, o1 = build()
    for _ in range(6):
        step(m1, o1, *batches[0])
    for b in batches:
        step(m1, o1, *b)
    probe = batches[1][1][:8] + m1.embedding.field_offsets
    ref = m1.embedding.variable.sparse_read(probe).clone()

    m2, o2 = build()
    static = tuple(t.clone() for t in batches[0])
    for _ in range(3):
        step(m2, o2, *static)
    torch.cuda.synchronize()
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            step(m2, o2, *static)
    torch.cuda.current_stream().wait_stream(side)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step(m2, o2, *static)   # must capture (reservation active)
    for b in batches:
        static[0].copy_(b[0])
        static[1].copy_(b[1])
        static[2].copy_(b[2])
        g.replay()
    torch.cuda.synchronize()
    got = m2.embedding.variable.sparse_read(probe)
    assert torch.allclose(got, ref, atol=1e-4, rtol=1e-4)
