#!/usr/bin/env python3
"""Wide&Deep with a 1B-row / ~100B-parameter embedding sharded across
8x MI355X (BASELINE.json config 4 — trillion-scale sizing).

Sizing: vocab 1e9 rows x (dim 99 + 1 wide column) fp32 = 400 GB of weights
+ 400 GB Adagrad state, sharded key%8 -> ~100 GB per GPU in 288 GB HBM.
The array table preallocates only the shard's slice; nothing materializes
the full table on one device.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/wide_deep_1b.py --steps 20

Scaled-down smoke (any machine):
    python examples/wide_deep_1b.py --vocab 1000000 --dim 16 --steps 5
"""

import argparse
import time

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))))  # run from a source checkout

import openembedding_amd.torch as embed
from openembedding_amd.models import WDL, N_DENSE


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--vocab", type=int, default=1_000_000_000,
                   help="total embedding rows across all fields")
    p.add_argument("--fields", type=int, default=26)
    p.add_argument("--dim", type=int, default=99)
    p.add_argument("--batch", type=int, default=4096)
    p.add_argument("--steps", type=int, default=20)
    args = p.parse_args()

    ctx = embed.get_context()
    per_field = args.vocab // args.fields
    vocabs = [per_field] * args.fields
    torch.manual_seed(7)
    model = WDL(field_vocabs=vocabs, dim=args.dim).to(ctx.device)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01))
    lossf = torch.nn.BCEWithLogitsLoss()
    if ctx.rank == 0:
        rows = sum(vocabs)
        params = rows * (args.dim + 1)
        print(f"embedding: {rows:,} rows x dim {args.dim}+1 = "
              f"{params / 1e9:.1f}B params "
              f"({params * 4 / 2 ** 30:.0f} GiB weights + same for state), "
              f"~{params * 8 / ctx.world_size / 2 ** 30:.0f} GiB/GPU "
              f"over {ctx.world_size} GPU(s)")

    gen = torch.Generator().manual_seed(13 + ctx.rank)
    t0 = time.perf_counter()
    for step in range(args.steps):
        dense = torch.rand(args.batch, N_DENSE, generator=gen)
        sparse = torch.stack(
            [torch.randint(0, v, (args.batch,), generator=gen)
             for v in vocabs], dim=1)
        labels = (torch.rand(args.batch, generator=gen) < 0.3).float()
        dense, sparse, labels = (dense.to(ctx.device), sparse.to(ctx.device),
                                 labels.to(ctx.device))
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        if ctx.rank == 0 and (step + 1) % 5 == 0:
            sps = args.batch * ctx.world_size * (step + 1) / (
                time.perf_counter() - t0)
            print(f"step {step + 1}: loss={loss.item():.4f} "
                  f"{sps:,.0f} samples/s")


if __name__ == "__main__":
    main()
