"""Metrics accumulators/reporter and the EnvConfig tree (SURVEY §5 parity:
reference VTIMER + accumulator report, WorkerContext.cpp:24-41,140-163;
EnvConfig defaults, client/EnvConfig.cpp:8-78; unknown-key warnings,
variable/Factory.h:64-76)."""

import io
import time
import warnings

import pytest
import torch

from openembedding_amd.config import EnvConfig
from openembedding_amd.utils.metrics import (
    MetricRegistry, REGISTRY, Reporter, set_perf, stage_timer)


def test_accumulator_stats():
    r = MetricRegistry()
    r.add("x", 1.0)
    r.add("x", 3.0)
    a = r.accumulator("x")
    assert a.n == 2 and a.total == 4.0 and a.vmin == 1.0 and a.vmax == 3.0
    assert a.mean == 2.0
    lines = r.report_lines(reset=True)
    assert len(lines) == 1 and lines[0].startswith("x: n=2")
    assert r.report_lines(reset=True) == []  # reset cleared it


def test_stage_timer_gated_on_perf():
    r = REGISTRY
    old = r.perf
    try:
        r.perf = False
        with stage_timer("g", "off"):
            pass
        assert r.accumulator("g.off_ms").n == 0
        set_perf(True)
        with stage_timer("g", "on"):
            time.sleep(0.01)
        a = r.accumulator("g.on_ms")
        assert a.n == 1 and a.total >= 5.0  # ms
    finally:
        r.perf = old


def test_reporter_prints_and_resets():
    r = MetricRegistry()
    r.add("pull_indices", 100)
    buf = io.StringIO()
    rep = Reporter(interval=0.01, rank=0, registry=r, out=buf)
    rep.report_once()
    assert "pull_indices" in buf.getvalue()
    assert r.report_lines() == []


def test_reporter_only_rank0():
    rep = Reporter(interval=0.01, rank=1, registry=MetricRegistry())
    rep.start()
    assert rep._thread is None


def test_pull_feeds_registry():
    from openembedding_amd.core.variable import VariableMeta, VariableShard
    from openembedding_amd.parallel.sharded import ShardedVariable

    v = ShardedVariable(VariableShard(
        VariableMeta(variable_id=901, embedding_dim=4, vocabulary_size=100)))
    v.set_initializer("constant", value=0.5)
    before = REGISTRY.accumulator("pull_indices").total
    out, _ = v.pull(torch.tensor([1, 2, 2, 3]))
    assert out.shape == (4, 4)
    assert REGISTRY.accumulator("pull_indices").total == before + 4
    assert v.stat_pull_indices == 4 and v.stat_pull_unique == 3


def test_env_config_defaults():
    c = EnvConfig.parse(None)
    assert c.server.report_interval == 0
    assert c.server.update_early_return is True
    assert c.server.message_compress == ""
    assert c.master.type == "tcp"


def test_env_config_yaml_and_json():
    y = "server:\n  report_interval: 5\n  cache_size: 1024\nmaster:\n  endpoint: 'a:1'\n"
    c = EnvConfig.parse(y)
    assert c.server.report_interval == 5
    assert c.server.cache_size_mb == 1024  # reference name cache_size aliased
    assert c.master.endpoint == "a:1"
    j = '{"server": {"update_early_return": false}}'
    assert EnvConfig.parse(j).server.update_early_return is False


def test_env_config_unknown_key_warns():
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        EnvConfig.parse("server:\n  bogus_knob: 1\n")
    assert any("bogus_knob" in str(x.message) for x in w)


def test_env_config_rejects_non_mapping():
    with pytest.raises(ValueError):
        EnvConfig.parse("[1,2]")


def test_context_parses_flags_config(tmp_path):
    import openembedding_amd as oe
    from openembedding_amd.context import Context

    old = oe.flags.config
    try:
        oe.flags.config = "server:\n  report_interval: 0\n"
        ctx = Context(device="cpu")
        assert ctx.config.server.report_interval == 0
        ctx.finalize()
    finally:
        oe.flags.config = old


def test_inject_patches_nn_embedding():
    # the reference's laboratory/inject demo: unmodified model code builds
    # PS-backed embeddings after install(); uninstall restores torch
    import torch
    import torch.nn as nn
    import openembedding_amd.inject as inject
    import openembedding_amd.torch as embed

    inject.install(sparse_as_dense_size=64)
    try:
        big = nn.Embedding(1000, 8)
        small = nn.Embedding(10, 8)
        hashed = nn.Embedding(-1, 8)
        padded = nn.Embedding(1000, 8, padding_idx=0)  # no PS equivalent
        assert isinstance(big, embed.Embedding)
        assert isinstance(hashed, embed.Embedding)
        assert type(small).__name__ == "Embedding" and not isinstance(
            small, embed.Embedding)
        assert isinstance(padded, inject._original)
        out = big(torch.tensor([[1, 2, 999]]))
        assert out.shape == (1, 3, 8)
    finally:
        inject.uninstall()
    plain = nn.Embedding(1000, 8)
    assert isinstance(plain, inject._original)
    assert nn.Embedding is inject._original


def test_top_level_parity_surface():
    # the reference's module-level API shape (openembedding/__init__.py):
    # flags, Master, Server, version — all present and minimally functional
    import openembedding_amd as oe
    assert oe.version == oe.__version__
    assert hasattr(oe.flags, "config")
    assert hasattr(oe.flags, "master_endpoint")
    assert hasattr(oe.flags, "num_workers")
    assert hasattr(oe.flags, "wait_num_servers")
    m = oe.Master(bind_ip="127.0.0.1", port=7777)
    assert m.running and m.endpoint == "127.0.0.1:7777"
    s = oe.Server(master_endpoint=m.endpoint)
    assert s.join() is None


def test_oeamd_device_env_override(monkeypatch):
    """OEAMD_DEVICE forces the context device (pairs with OEAMD_BACKEND
    for the N-ranks-on-1-GPU gloo rehearsal — see
    profiles/final_validation_r2.md)."""
    import openembedding_amd.context as cm
    if cm._context is not None:
        cm._context.finalize()
        cm._context = None
    import openembedding_amd.torch as api
    api._tracked.clear()
    monkeypatch.setenv("OEAMD_DEVICE", "cpu")
    ctx = cm.Context()
    assert ctx.device.type == "cpu"
    ctx.finalize()
    cm._context = None
