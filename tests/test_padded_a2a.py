"""Padded (fixed-capacity, sync-free) all-to-all route tests.

The padded route ships [world, cap] key blocks padded with the reserved key
-1 instead of exact variable splits (parallel/sharded.py _pull_remote_padded;
wire format in ops/csrc/embops.hip k_bucketize_pad). These tests check, on
the CPU engine:

  - forced-remote world-1: padded == exact (pull values + post-commit state)
  - gloo world-2/3: padded multi-rank == world-1 exact reference
  - overflow is detected loudly, never silent zeros
  - plan capacity policy

GPU equivalence of the HIP kernels runs in tests/test_gpu_padded.py.
"""

import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

DIM = 4
VOCAB = 1000


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _init(rank, world, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")


def _clean_env():
    for v in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        os.environ.pop(v, None)


def _rank_batches(rank):
    g = torch.Generator().manual_seed(100 + rank)
    keys = torch.randint(0, VOCAB, (64,), generator=g, dtype=torch.int64)
    grads = torch.randn(64, DIM, generator=g)
    return keys, grads


def _make_var(ctx):
    st = ctx.create_storage()
    var = st.create_variable(VOCAB, DIM)
    var.set_initializer("uniform", minval=-1, maxval=1)
    var.set_optimizer("test")
    return st, var


def test_forced_remote_padded_matches_exact():
    """world-1, collectives are identity: the padded bucketize/scatter
    pipeline must reproduce the exact path bit-for-bit (CPU fallback)."""
    from openembedding_amd.context import Context

    results = []
    for padded in (False, True):
        ctx = Context(device="cpu")
        st, var = _make_var(ctx)
        var._force_remote = True
        var._padded = padded
        keys = torch.tensor([5, 7, 5, 11, 7, 7, 999, 0], dtype=torch.int64)
        grads = torch.arange(8 * DIM, dtype=torch.float32).reshape(8, DIM)
        out, h = var.pull(keys)
        var.push(h, grads)
        st.update_weights()
        after, _ = var.pull(keys, readonly=True)
        var.check_padded_overflow()
        results.append((out, after))
    torch.testing.assert_close(results[0][0], results[1][0])
    torch.testing.assert_close(results[0][1], results[1][1])


def _worker_padded(rank, world, port, tmp):
    from openembedding_amd.context import Context
    from openembedding_amd.parallel.sharded import ShardedVariable

    _init(rank, world, port)
    ShardedVariable._padded = True
    try:
        ctx = Context(device="cpu")
        st, var = _make_var(ctx)
        keys, grads = _rank_batches(rank)
        out, h = var.pull(keys)
        var.push(h, grads)
        st.update_weights()
        out2, _ = var.pull(keys, readonly=True)
        var.check_padded_overflow()
        torch.save({"keys": keys, "out": out, "out2": out2},
                   os.path.join(tmp, f"result_{rank}.pt"))
        dist.barrier()
    finally:
        ShardedVariable._padded = None
        dist.destroy_process_group()


def _reference_world1(rank_ids):
    from openembedding_amd.context import Context

    ctx = Context(device="cpu")
    st, var = _make_var(ctx)
    handles = []
    for r in rank_ids:
        k, g = _rank_batches(r)
        out, h = var.pull(k)
        handles.append((h, g))
    for h, g in handles:
        var.push(h, g)
    st.update_weights()
    return var


@pytest.mark.timeout(240)
@pytest.mark.parametrize("world", [2, 3])
def test_padded_world_matches_world1(world, tmp_path):
    port = _free_port()
    mp.spawn(_worker_padded, args=(world, port, str(tmp_path)), nprocs=world,
             join=True)
    _clean_env()
    var = _reference_world1(range(world))
    for r in range(world):
        res = torch.load(tmp_path / f"result_{r}.pt", weights_only=True)
        after, _ = var.pull(res["keys"], readonly=True)
        torch.testing.assert_close(res["out2"], after, rtol=1e-5, atol=1e-5)


def _worker_model_padded(rank, world, port, tmp):
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch
    from openembedding_amd.parallel.sharded import ShardedVariable

    _init(rank, world, port)
    ShardedVariable._padded = True
    try:
        torch.manual_seed(0)
        fv = [50, 3, 1000, 40] + [100] * 22
        model = DeepFM(field_vocabs=fv, dim=4)
        opt = embed.distributed_optimizer(
            torch.optim.Adagrad(model.parameters(), lr=0.01))
        lossf = torch.nn.BCEWithLogitsLoss()
        g = torch.Generator().manual_seed(10 + rank)
        for step in range(3):
            dense, sparse, labels = synthetic_batch(64, field_vocabs=fv,
                                                    generator=g)
            opt.zero_grad()
            loss = lossf(model(dense, sparse), labels)
            loss.backward()
            opt.step()
        for v in embed.get_context().variables.values():
            v.check_padded_overflow()
        p = torch.cat([q.detach().reshape(-1)
                       for q in model.dnn.parameters()])
        gathered = [torch.empty_like(p) for _ in range(world)]
        dist.all_gather(gathered, p)
        assert torch.allclose(gathered[0], gathered[1], rtol=1e-6, atol=1e-6)
        dist.barrier()
    finally:
        ShardedVariable._padded = None
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_deepfm_world2_padded(tmp_path):
    """Full model step over the padded route (prefix/pipeline unchanged)."""
    port = _free_port()
    mp.spawn(_worker_model_padded, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)
    _clean_env()


def test_overflow_detected(monkeypatch):
    """cap smaller than the per-owner unique count: the dropped keys must
    surface as a RuntimeError at the overflow check, not as silent zeros."""
    from openembedding_amd.context import Context

    monkeypatch.setenv("OEAMD_A2A_SLACK", "0.01")
    ctx = Context(device="cpu")
    st = ctx.create_storage()
    var = st.create_variable(1 << 20, DIM)
    var.set_initializer("constant", value=1.0)
    var.set_optimizer("test")
    var._force_remote = True
    var._padded = True
    keys = torch.arange(2048, dtype=torch.int64)  # all unique, one owner
    out, h = var.pull(keys)
    with pytest.raises(RuntimeError, match="overflow"):
        var.check_padded_overflow()
    # and the overflowed positions read zeros (defined, not garbage)
    assert float(out[-1].abs().sum()) == 0.0


def test_plan_capacity_policy():
    from openembedding_amd.parallel.sharded import _PaddedPlan

    p = _PaddedPlan(8, 106496, torch.device("cpu"))
    # cap rounds ceil(n*slack/world) up to 256 and never exceeds n
    assert p.cap % 256 == 0 and p.cap < 106496
    assert p.cap * 8 >= 2 * 106496  # slack 2 headroom over even spread
    small = _PaddedPlan(8, 64, torch.device("cpu"))
    assert small.cap == 64  # tiny batches: cap == n (overflow impossible)
    assert small.send_keys.numel() == 8 * 64


def test_reserved_key_zero_row():
    """key -1 through the padded route: defined zeros (GPU contract),
    matching the round-1 reservation."""
    from openembedding_amd.context import Context

    ctx = Context(device="cpu")
    st, var = _make_var(ctx)
    var._force_remote = True
    var._padded = True
    keys = torch.tensor([3, -1, 5], dtype=torch.int64)
    out, h = var.pull(keys)
    assert float(out[1].abs().sum()) == 0.0
    assert float(out[0].abs().sum()) > 0.0


@pytest.mark.timeout(240)
def test_padded_fuzz_vs_exact():
    """Property fuzz of the padded route (CPU fallback, forced-remote):
    random key mixes — duplicates, hot keys, the reserved -1, shapes not
    divisible by anything — must match the exact route bit-for-bit."""
    from openembedding_amd.context import Context

    g = torch.Generator().manual_seed(42)
    for trial in range(10):
        n = int(torch.randint(1, 300, (1,), generator=g))
        hot = int(torch.randint(1, 10, (1,), generator=g))
        keys = torch.randint(0, 50, (n,), generator=g, dtype=torch.int64)
        keys[torch.randint(0, n, (min(hot, n),), generator=g)] = 7
        grads = torch.randn(n, DIM, generator=g)
        results = []
        for padded in (False, True):
            ctx = Context(device="cpu")
            st, var = _make_var(ctx)
            var._force_remote = True
            var._padded = padded
            out, h = var.pull(keys)
            var.push(h, grads)
            st.update_weights()
            after, _ = var.pull(keys, readonly=True)
            var.check_padded_overflow()
            results.append((out, after))
        torch.testing.assert_close(results[0][0], results[1][0])
        torch.testing.assert_close(results[0][1], results[1][1])


@pytest.mark.timeout(240)
def test_padded_two_outstanding_pulls():
    """Two pulls before their pushes (the prefetch pattern): the plan
    rotation must keep both handles' wire maps alive."""
    from openembedding_amd.context import Context

    ctx = Context(device="cpu")
    st, var = _make_var(ctx)
    var._force_remote = True
    var._padded = True
    k1 = torch.tensor([3, 5, 7, 5], dtype=torch.int64)
    k2 = torch.tensor([5, 9, 11, 3], dtype=torch.int64)
    out1, h1 = var.pull(k1)
    out2, h2 = var.pull(k2)        # same shape -> would reuse h1's plan
    assert h1.plan is not h2.plan, "outstanding pulls must not share a plan"
    var.push(h1, torch.ones_like(out1))
    var.push(h2, torch.full_like(out2, 2.0))
    st.update_weights()

    # reference semantics: one optimizer step per unique key over summed
    # grads — compare with the exact path doing the same two-pull commit
    ctx2 = Context(device="cpu")
    st2, var2 = _make_var(ctx2)
    o1, g1 = var2.pull(k1)
    o2, g2 = var2.pull(k2)
    var2.push(g1, torch.ones_like(o1))
    var2.push(g2, torch.full_like(o2, 2.0))
    st2.update_weights()
    probe = torch.tensor([3, 5, 7, 9, 11], dtype=torch.int64)
    a, _ = var.pull(probe, readonly=True)
    b, _ = var2.pull(probe, readonly=True)
    torch.testing.assert_close(a, b)
