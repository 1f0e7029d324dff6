"""persist/should_persist/restore API over the capacity tier (reference
pmem trio exb.py:697-705 + pmem_c_api_test.cpp training loop with periodic
persist)."""

import torch

import openembedding_amd as oe
import openembedding_amd.torch as embed


def test_persist_cycle_with_tiered_variable(tmp_path):
    old = oe.flags.config
    oe.flags.config = "server:\n  cache_size: 1\n"  # ~1 MiB cache -> tiny
    try:
        ctx = embed.get_context(device="cpu")
        # hash-mode variable -> tiered shard under cache_size
        emb = embed.Embedding(-1, 4)
        from openembedding_amd.core.tiered import TieredVariableShard
        assert isinstance(emb.variable.sharded.shard, TieredVariableShard)
        emb.variable.sharded.shard.cache_rows = 8  # force pressure
        opt = embed.distributed_optimizer(
            torch.optim.Adagrad([torch.nn.Parameter(torch.zeros(1))], lr=0.1))
        gen = torch.Generator().manual_seed(0)
        for step in range(6):
            keys = torch.randint(0, 64, (16,), generator=gen,
                                 dtype=torch.int64)
            out = emb(keys)
            out.sum().backward()
            opt.step()
        assert embed.should_persist_server_model()
        uri = str(tmp_path / "persist")
        embed.persist_server_model(uri)
        assert not embed.should_persist_server_model()
        ref = emb.variable.sparse_read(torch.arange(64))
        embed.restore_server_model(uri)
        got = emb.variable.sparse_read(torch.arange(64))
        assert torch.allclose(ref, got)
    finally:
        oe.flags.config = old


def test_untired_should_persist_false():
    embed.get_context(device="cpu")
    emb = embed.Embedding(100, 4)
    _ = emb(torch.tensor([1, 2]))
    assert not embed.should_persist_server_model()
