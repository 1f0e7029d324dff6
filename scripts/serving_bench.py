#!/usr/bin/env python3
"""Serving read-path microbenchmark: batched pulls/s from a loaded model
(GPU HIP tables or CPU), the serving-side analogue of the training bench."""

import os
import sys
import tempfile
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import torch  # noqa: E402


def main():
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch
    from openembedding_amd.serving import ModelController

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    model = DeepFM(dim=9).to(dev)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01))
    lossf = torch.nn.BCEWithLogitsLoss()
    for _ in range(3):
        dense, sparse, labels = synthetic_batch(4096, device=dev)
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
    uri = tempfile.mkdtemp(prefix="oe_srv_bench_")
    embed.save_server_model(uri)
    ctx = embed.get_context()
    sign = f"{ctx.model_uuid}-{ctx.model_version}"
    c = ModelController(device=dev)
    c.create_model(uri)
    var = c.manager.find_model_variable(sign, 0)

    batch, steps, warmup = 4096, 200, 20
    emb = model.embedding
    gen = torch.Generator().manual_seed(9)
    probes = []
    for _ in range(8):
        _, sp, _ = synthetic_batch(batch, generator=gen)
        probes.append((sp.to(dev) + emb.field_offsets))
    for i in range(warmup):
        var.pull_weights(probes[i % 8])
    if dev.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        var.pull_weights(probes[i % 8])
    if dev.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    n = batch * 26
    print(f"serving pull: {dev} batch {batch}x26 keys: "
          f"{steps / dt:,.0f} pulls/s, {steps * n / dt / 1e6:,.1f}M "
          f"keys/s, {dt / steps * 1e6:.0f} us/pull "
          f"(shard={type(var.shard).__name__})")


if __name__ == "__main__":
    main()
