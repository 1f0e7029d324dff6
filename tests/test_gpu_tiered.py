"""GPU capacity tier (HBM cache over pinned host DRAM): equivalence with
the untired HIP shard under eviction pressure, fault-back, persistence
(BASELINE.json config 5; reference PMem tier semantics)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _mk(cache_rows=32, dim=8):
    from openembedding_amd.core.tiered_gpu import HipTieredVariableShard
    from openembedding_amd.core.variable import (HASH_VOCAB_THRESHOLD,
                                                 VariableMeta)
    from openembedding_amd.core.variable_gpu import HipVariableShard

    meta = VariableMeta(variable_id=7, embedding_dim=dim,
                        vocabulary_size=HASH_VOCAB_THRESHOLD)
    t = HipTieredVariableShard(meta, device=DEV, cache_rows=cache_rows)
    r = HipVariableShard(meta, device=DEV)
    for sh in (t, r):
        sh.set_initializer("uniform", minval=-1.0, maxval=1.0)
        sh.set_optimizer("adagrad", learning_rate=0.1)
    return t, r


def _step(sh, keys):
    uk, inv = torch.unique(keys, return_inverse=True)
    w = sh.pull(uk)
    g = torch.ones((uk.numel(), w.shape[1]), device=DEV)
    c = torch.zeros(uk.numel(), dtype=torch.int64, device=DEV)
    c.index_add_(0, inv, torch.ones_like(inv))
    sh.push(uk, g, c)
    sh.update_weights()
    return uk, w


def test_equivalence_under_eviction():
    t, r = _mk(cache_rows=32)
    gen = torch.Generator().manual_seed(5)
    for step in range(20):
        keys = torch.randint(0, 256, (48,), generator=gen,
                             dtype=torch.int64).to(DEV)
        ukt, wt = _step(t, keys)
        ukr, wr = _step(r, keys)
        assert torch.equal(wt, wr), f"diverged at step {step}"
    assert len(t._host_of) > 0, "no eviction happened — raise pressure"
    allk = torch.arange(256, dtype=torch.int64, device=DEV)
    assert torch.equal(t.pull_readonly(allk), r.pull_readonly(allk))
    assert t.num_rows == r.num_rows


def test_fault_back_roundtrip():
    t, _ = _mk(cache_rows=8)
    k1 = torch.arange(0, 8, dtype=torch.int64, device=DEV)
    w1 = t.pull(k1).clone()
    t.pull(torch.arange(8, 16, dtype=torch.int64, device=DEV))
    assert len(t._host_of) > 0
    assert torch.equal(t.pull(k1), w1)


def test_persist_and_export():
    t, r = _mk(cache_rows=16)
    keys = torch.arange(0, 24, dtype=torch.int64, device=DEV)
    _step(t, keys)
    _step(r, keys)
    assert t.should_persist()
    t.persist()
    assert not t.should_persist()
    kt, wt, st = t.export_rows()
    kr, wr, sr = r.export_rows()
    ot, orr = torch.argsort(kt), torch.argsort(kr)
    assert torch.equal(kt[ot], kr[orr])
    assert torch.equal(wt[ot], wr[orr])
    assert torch.equal(st[ot], sr[orr])


def test_sharded_engine_routes_exact_path():
    """The engine must not take the bounded path for a tiered shard."""
    from openembedding_amd.parallel.sharded import ShardedVariable

    t, _ = _mk(cache_rows=16)
    v = ShardedVariable(t)
    keys = torch.randint(0, 64, (40,), dtype=torch.int64).to(DEV)
    out, h = v.pull(keys)
    assert not h.bounded
    v.push(h, torch.ones_like(out))
    v.update_weights()
    torch.cuda.synchronize()
