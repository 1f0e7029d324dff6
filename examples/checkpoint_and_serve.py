#!/usr/bin/env python3
"""Train -> dump -> serve round trip (the reference's TF-Serving flow,
examples/run/criteo_deepctr_restful.sh, without TF):

1. trains DeepFM a few steps;
2. dumps the server model (reference dump format, checkpoint.py);
3. loads it into the serving ModelController and answers REST pulls;
4. checks served rows equal the trained rows.

Run: python examples/checkpoint_and_serve.py
"""

import tempfile

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))))  # run from a source checkout

import openembedding_amd.torch as embed
from openembedding_amd.models import DeepFM, synthetic_batch
from openembedding_amd.serving import ModelController, make_app


def main():
    ctx = embed.get_context()
    torch.manual_seed(0)
    # small per-field vocabularies keep the SavedModel materialization
    # below quick (full Criteo cardinalities work the same, just bigger)
    fv = [1000] * 26
    model = DeepFM(field_vocabs=fv, dim=9).to(ctx.device)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01))
    lossf = torch.nn.BCEWithLogitsLoss()
    for _ in range(5):
        dense, sparse, labels = synthetic_batch(
            256, field_vocabs=fv, device=str(ctx.device))
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
    print(f"trained: loss={loss.item():.4f}")

    uri = tempfile.mkdtemp(prefix="oe_dump_")
    embed.save_server_model(uri)
    sign = f"{ctx.model_uuid}-{ctx.model_version}"
    print(f"dumped to {uri} sign={sign}")

    controller = ModelController()
    controller.create_model(uri)

    # REST round-trip via the in-process test client
    from fastapi.testclient import TestClient
    client = TestClient(make_app(controller))
    print("models:", [m["model_sign"] for m in client.get("/models").json()])

    # compare a few served rows to the live trained table
    emb = model.embedding
    probe = sparse[:4] + emb.field_offsets          # global keys, [4, 26]
    live = emb.variable.sparse_read(probe)
    r = client.post(f"/models/{sign}/variables/0/pull",
                    json={"indices": probe.cpu().tolist()})
    served = torch.tensor(r.json()["weights"], device=live.device)
    assert torch.allclose(live, served, atol=1e-6), "served rows differ!"
    print("serving matches training: OK")

    # TF-Serving-loadable SavedModel export (the reference's
    # save_as_original_model contract): a standalone directory TF loads
    # with no OpenEmbedding runtime
    sm = tempfile.mkdtemp(prefix="oe_savedmodel_")
    embed.save_as_original_model(model, sm)
    import os
    assert os.path.exists(os.path.join(sm, "saved_model.pb"))
    print(f"SavedModel exported to {sm} "
          f"(saved_model.pb + variables bundle)")


if __name__ == "__main__":
    main()
