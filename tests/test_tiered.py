"""Capacity tier: device row-cache over host backing store
(reference PMem tier tests, variable/pmem_embedding_table_test.cpp:
cache/evict pressure with a tiny cache, checkpoint watermark, reload;
pmem_c_api_test.cpp persist/restore loop)."""

import pytest
import torch

from openembedding_amd.core.tiered import TieredVariableShard
from openembedding_amd.core.variable import (
    HASH_VOCAB_THRESHOLD, VariableMeta, VariableShard)


def _meta(vid=0, dim=4):
    return VariableMeta(variable_id=vid, embedding_dim=dim,
                        vocabulary_size=HASH_VOCAB_THRESHOLD)


def _mk(cache_rows=8, dim=4, opt=("adagrad", {"learning_rate": 0.1})):
    t = TieredVariableShard(_meta(dim=dim), cache_rows=cache_rows)
    r = VariableShard(_meta(dim=dim))
    for sh in (t, r):
        sh.set_initializer("uniform", minval=-1.0, maxval=1.0)
        sh.set_optimizer(opt[0], **opt[1])
    return t, r


def _step(sh, keys):
    w = sh.pull(keys)
    g = torch.ones_like(w) * 0.5
    u, inv = torch.unique(keys, return_inverse=True)
    ug = torch.zeros((u.numel(), w.shape[1]))
    ug.index_add_(0, inv, g)
    c = torch.zeros(u.numel(), dtype=torch.int64)
    c.index_add_(0, inv, torch.ones_like(inv))
    sh.push(u, ug, c)
    sh.update_weights()
    return w


def test_requires_hash_mode():
    with pytest.raises(ValueError):
        TieredVariableShard(VariableMeta(variable_id=0, embedding_dim=4,
                                         vocabulary_size=100))


def test_equivalence_under_eviction_pressure():
    """Tiny cache (8 rows), 64 keys revisited over 30 steps: weights must
    bit-match an untired shard (the tier only moves rows, never changes
    math) — the reference's mirror-check approach (c_api_test.h:123-133)."""
    t, r = _mk(cache_rows=8)
    gen = torch.Generator().manual_seed(3)
    for step in range(30):
        keys = torch.randint(0, 64, (16,), generator=gen, dtype=torch.int64)
        wt = _step(t, keys)
        wr = _step(r, keys)
        assert torch.equal(wt, wr), f"diverged at step {step}"
    assert t.num_rows == r.num_rows
    # every key readable after all that churn
    all_keys = torch.arange(64, dtype=torch.int64)
    assert torch.equal(t.pull_readonly(all_keys), r.pull_readonly(all_keys))
    # cache stayed bounded (one batch may overshoot transiently)
    assert t._nrows <= 8 + 16


def test_rows_spill_and_fault_back():
    t, _ = _mk(cache_rows=4)
    k1 = torch.arange(0, 4, dtype=torch.int64)
    k2 = torch.arange(4, 8, dtype=torch.int64)
    w1 = t.pull(k1).clone()
    _ = t.pull(k2)  # evicts some of k1
    assert len(t._host_of) > 0
    w1b = t.pull(k1)  # faults them back
    assert torch.equal(w1, w1b)


def test_readonly_serves_host_rows_without_promotion():
    t, _ = _mk(cache_rows=4)
    k1 = torch.arange(0, 4, dtype=torch.int64)
    w1 = t.pull(k1).clone()
    t.pull(torch.arange(4, 8, dtype=torch.int64))
    spilled = [k for k in range(4) if k in t._host_of]
    assert spilled
    sk = torch.tensor(spilled, dtype=torch.int64)
    got = t.pull_readonly(sk)
    assert torch.equal(got, w1[sk])
    assert all(k in t._host_of for k in spilled)  # not promoted


def test_should_persist_signal_and_watermark():
    t, _ = _mk(cache_rows=4)
    assert not t.should_persist()
    _step(t, torch.arange(0, 4, dtype=torch.int64))
    assert not t.should_persist()  # not full yet
    _step(t, torch.arange(4, 10, dtype=torch.int64))
    assert t.should_persist()      # cache overflowed since last ckpt
    wid = t.persist()
    assert wid == t.work_id
    assert not t.should_persist()  # pending checkpoint blocks the signal
    t.checkpoint_committed()
    assert not t.should_persist()  # nothing new since persist
    _step(t, torch.arange(20, 30, dtype=torch.int64))
    assert t.should_persist()      # refilled after commit


def test_persist_makes_host_image_complete():
    t, r = _mk(cache_rows=8)
    keys = torch.arange(0, 12, dtype=torch.int64)
    _step(t, keys)
    _step(r, keys)
    t.persist()
    # host image covers every row
    assert len(t._host_of) == t.num_rows
    # export matches the untired shard (sorted by key)
    kt, wt, st = t.export_rows()
    kr, wr, sr = r.export_rows()
    ot, orr = torch.argsort(kt), torch.argsort(kr)
    assert torch.equal(kt[ot], kr[orr])
    assert torch.equal(wt[ot], wr[orr])
    assert torch.equal(st[ot], sr[orr])


def test_export_import_roundtrip_through_tier():
    t, _ = _mk(cache_rows=4)
    keys = torch.arange(0, 10, dtype=torch.int64)
    _step(t, keys)
    k, w, s = t.export_rows()
    fresh = TieredVariableShard(_meta(vid=1), cache_rows=4)
    fresh.set_initializer("constant", value=0.0)
    fresh.set_optimizer("adagrad", learning_rate=0.1)
    fresh.import_rows(k, w, s)
    assert torch.equal(fresh.pull_readonly(keys), t.pull_readonly(keys))


def test_clear_resets_both_tiers():
    t, _ = _mk(cache_rows=4)
    _step(t, torch.arange(0, 10, dtype=torch.int64))
    assert t.num_rows == 10
    t.clear()
    assert t.num_rows == 0 and not t._host_of
    assert torch.equal(t.pull_readonly(torch.tensor([1])),
                       torch.zeros(1, 4))


def test_tiered_world2_matches_untiered(tmp_path):
    """CPU capacity tier at world 2 (gloo): a tiered run over the sharded
    route must match an untiered one bit-for-bit (the tier is a storage
    policy, not a semantics change)."""
    import os
    import socket

    import torch.distributed as dist
    import torch.multiprocessing as mp

    def _free_port():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    port = _free_port()
    mp.spawn(_tier_worker, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)
    for v in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        os.environ.pop(v, None)
    a = torch.load(tmp_path / "after_tiered_0.pt", weights_only=True)
    b = torch.load(tmp_path / "after_plain_0.pt", weights_only=True)
    torch.testing.assert_close(a, b)


def _tier_worker(rank, world, port, tmp):
    import os

    import torch.distributed as dist

    import openembedding_amd as oe
    from openembedding_amd.context import Context

    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            init_method=f"tcp://127.0.0.1:{port}")
    g = torch.Generator().manual_seed(31)
    batches = [(torch.randint(0, 40000, (600,), generator=g,
                              dtype=torch.int64),
                torch.randn(600, 4, generator=g)) for _ in range(6)]
    # ~3.4k unique keys vs the 1024-row cache floor -> real evictions
    probe = torch.unique(torch.cat([k for k, _ in batches]))
    results = {}
    for label, cfg in (("tiered", "server:\n  cache_size: 1\n"),
                       ("plain", "")):
        old = oe.flags.config
        oe.flags.config = cfg
        try:
            ctx = Context(device="cpu")
            st = ctx.create_storage()
            var = st.create_variable(1 << 63, 4)   # hash mode (tier req.)
            var.set_initializer("uniform", minval=-1, maxval=1)
            var.set_optimizer("adagrad", learning_rate=0.1)
            for keys, grads in batches:
                out, h = var.pull(keys)
                var.push(h, grads)
                st.update_weights()
            after, _ = var.pull(probe, readonly=True)
            results[label] = after
        finally:
            oe.flags.config = old
    if rank == 0:
        torch.save(results["tiered"], os.path.join(tmp, "after_tiered_0.pt"))
        torch.save(results["plain"], os.path.join(tmp, "after_plain_0.pt"))
    dist.barrier()
    dist.destroy_process_group()
