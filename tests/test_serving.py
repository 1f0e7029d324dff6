"""Serving stack: ModelManager / ModelController / REST controller
(reference client/ModelController.cpp, entry/controller.cc; read-only pull
semantics EmbeddingPullOperator.cpp:179-181)."""

import os
import time

import pytest
import torch

import openembedding_amd.torch as embed
from openembedding_amd.serving import (
    ModelController, ModelStatus, make_app)


@pytest.fixture()
def dumped_model(tmp_path):
    """Train a tiny model 2 steps, dump it, return (uri, model, sign)."""
    torch.manual_seed(7)
    emb = embed.Embedding(50, 4)
    emb.variable.set_optimizer("adagrad", learning_rate=0.1)
    idx = torch.tensor([1, 2, 3, 17])
    out = emb(idx)
    out.sum().backward()
    embed.get_context().update_all_weights()
    uri = str(tmp_path / "dump")
    embed.save_server_model(uri)
    ctx = embed.get_context()
    sign = f"{ctx.model_uuid}-{ctx.model_version}"
    expect = emb.variable.sparse_read(idx).detach().clone()
    return uri, sign, idx, expect


def test_create_and_pull(dumped_model):
    uri, sign, idx, expect = dumped_model
    c = ModelController()
    m = c.create_model(uri)
    assert m.sign == sign and m.status == ModelStatus.NORMAL
    var = c.manager.find_model_variable(sign, 0)
    got = var.pull_weights(idx)
    assert torch.allclose(got, expect)
    # missing keys -> zeros (reference read-only get_weights)
    zeros = var.pull_weights(torch.tensor([44]))
    assert torch.equal(zeros, torch.zeros(1, 4))


def test_async_create(dumped_model):
    uri, sign, idx, expect = dumped_model
    c = ModelController()
    m = c.create_model(uri, wait=False)
    for _ in range(100):
        if m.status == ModelStatus.NORMAL:
            break
        time.sleep(0.05)
    assert m.status == ModelStatus.NORMAL


def test_duplicate_create_rejected(dumped_model):
    uri, sign, *_ = dumped_model
    c = ModelController()
    c.create_model(uri)
    with pytest.raises(ValueError):
        c.create_model(uri)


def test_delete_and_missing(dumped_model):
    uri, sign, *_ = dumped_model
    c = ModelController()
    c.create_model(uri)
    c.delete_model(sign)
    with pytest.raises(KeyError):
        c.manager.find_model_variable(sign, 0)
    with pytest.raises(KeyError):
        c.delete_model(sign)


def test_show_models_nodes(dumped_model):
    uri, sign, *_ = dumped_model
    c = ModelController()
    c.create_model(uri, sign="alias")
    models = c.show_models()
    assert [m["model_sign"] for m in models] == ["alias"]
    assert models[0]["variables"][0]["embedding_dim"] == 4
    nodes = c.show_nodes()
    assert nodes[0]["models"] == ["alias"]


def test_rest_api(dumped_model):
    from fastapi.testclient import TestClient

    uri, sign, idx, expect = dumped_model
    app = make_app()
    client = TestClient(app)
    r = client.post("/models", json={"model_uri": uri})
    assert r.status_code == 200, r.text
    assert r.json()["model_sign"] == sign
    # conflict on duplicate
    assert client.post("/models", json={"model_uri": uri}).status_code == 409
    # bad uri
    assert client.post("/models", json={"model_uri": uri + "x"}).status_code == 400
    assert client.get("/models").json()[0]["status"] == "NORMAL"
    assert client.get(f"/models/{sign}").status_code == 200
    assert client.get("/models/nope").status_code == 404
    r = client.post(f"/models/{sign}/variables/0/pull",
                    json={"indices": idx.tolist()})
    assert r.status_code == 200
    got = torch.tensor(r.json()["weights"])
    assert torch.allclose(got, expect, atol=1e-6)
    assert client.get("/nodes").json()[0]["node_id"] == 0
    assert client.delete(f"/models/{sign}").status_code == 200
    assert client.get(f"/models/{sign}").status_code == 404
    assert client.delete("/nodes/0").json()["shutdown_requested"] == 0


def test_metrics_endpoint(dumped_model):
    from fastapi.testclient import TestClient
    from openembedding_amd.utils.metrics import REGISTRY

    uri, sign, *_ = dumped_model
    c = ModelController()
    c.create_model(uri)
    REGISTRY.add("pull_indices", 7)
    client = TestClient(make_app(c))
    body = client.get("/metrics").text
    assert "openembedding_metric" in body
    assert "openembedding_models 1" in body


@pytest.mark.timeout(120)
def test_cli_server_boots(dumped_model, tmp_path):
    # `python -m openembedding_amd.serving --model-uri U` must boot, serve
    # the model list over real HTTP and die cleanly (reference controller
    # daemon entry, entry/controller.cc main)
    import socket
    import subprocess
    import sys
    import time as _t
    import urllib.request

    uri, sign, _idx, _expect = dumped_model
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.Popen(
        [sys.executable, "-m", "openembedding_amd.serving",
         "--host", "127.0.0.1", "--port", str(port),
         "--device", "cpu", "--model-uri", uri],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        body = None
        for _ in range(100):
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/models", timeout=2) as r:
                    body = r.read().decode()
                break
            except Exception:
                assert proc.poll() is None, (
                    "server died: " + proc.stdout.read().decode()[-2000:])
                _t.sleep(0.2)
        assert body is not None and sign in body
    finally:
        proc.terminate()
        proc.wait(timeout=20)
