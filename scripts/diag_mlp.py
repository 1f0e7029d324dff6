"""Diagnose the fused-MLP NaN: run identical steps through the fused path
and the plain bf16 torch path on the SAME initial weights, compare grads
and weights per step (GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import openembedding_amd.torch as embed
from openembedding_amd.models import DeepFM, synthetic_batch
from openembedding_amd.models.ctr import convert_mlp_bf16

DEV = "cuda:0"
torch.manual_seed(0)


def build(fused):
    import openembedding_amd.context as cm
    import openembedding_amd.torch as api
    if cm._context is not None:
        cm._context.finalize()
        cm._context = None
    api._tracked.clear()
    torch.manual_seed(0)
    m = convert_mlp_bf16(DeepFM(dim=9).to(DEV))
    if not fused:
        m.fused_mlp = False
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(m.parameters(), lr=0.01), flatten_dense=True)
    return m, opt


def run(m, opt, batches):
    lossf = torch.nn.BCEWithLogitsLoss()
    stats = []
    for dense, sparse, labels in batches:
        opt.zero_grad()
        loss = lossf(m(dense, sparse), labels)
        loss.backward()
        row = {"loss": loss.item()}
        for name, p in m.named_parameters():
            if p.grad is not None and p.numel():
                row[name] = (p.grad.float().norm().item(),
                             p.data.float().norm().item())
        opt.step()
        stats.append(row)
    return stats


gen = torch.Generator().manual_seed(7)
batches = [tuple(t.to(DEV) for t in synthetic_batch(2048, generator=gen))
           for _ in range(4)]

mf, of = build(True)
sf = run(mf, of, batches)
mp, op = build(False)
sp = run(mp, op, batches)

for i, (a, b) in enumerate(zip(sf, sp)):
    print(f"step {i}: fused loss={a['loss']:.4f} plain loss={b['loss']:.4f}")
    for k in sorted(a):
        if k == "loss":
            continue
        ga, wa = a[k]
        gb, wb = b.get(k, (float('nan'), float('nan')))
        flag = " <<<" if (abs(ga - gb) > 0.2 * (abs(gb) + 1e-3)) else ""
        print(f"  {k:28s} grad {ga:12.4f} vs {gb:12.4f} | w {wa:10.3f} vs {wb:10.3f}{flag}")
