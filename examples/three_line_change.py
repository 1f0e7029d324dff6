#!/usr/bin/env python3
"""The reference's selling point — convert an EXISTING torch CTR model with
a 3-line change (reference README + examples/criteo_deepctr_hook.py):

    model = embed.distributed_model(model)            # swap nn.Embedding -> PS
    opt = embed.distributed_optimizer(torch.optim.Adagrad(...))
    ... train exactly as before ...

``distributed_model`` walks the module tree and replaces every nn.Embedding
with a PS-backed one (tables smaller than sparse_as_dense_size stay
replicated dense — the reference's "cache" policy).
"""

import torch
import torch.nn as nn

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))))  # run from a source checkout

import openembedding_amd.torch as embed


class VanillaCTR(nn.Module):
    """A plain torch model someone wrote without openembedding_amd."""

    def __init__(self):
        super().__init__()
        self.user_emb = nn.Embedding(1_000_000, 16)   # big -> PS-backed
        self.item_emb = nn.Embedding(500_000, 16)     # big -> PS-backed
        self.country_emb = nn.Embedding(32, 16)       # small -> stays dense
        self.mlp = nn.Sequential(nn.Linear(48, 64), nn.ReLU(),
                                 nn.Linear(64, 1))

    def forward(self, user, item, country):
        x = torch.cat([self.user_emb(user), self.item_emb(item),
                       self.country_emb(country)], dim=1)
        return self.mlp(x).squeeze(-1)


def main():
    ctx = embed.get_context()
    model = VanillaCTR().to(ctx.device)

    # --- the 3-line change ---------------------------------------------
    model = embed.distributed_model(model, sparse_as_dense_size=64)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01))
    # --------------------------------------------------------------------

    assert isinstance(model.user_emb, embed.Embedding)
    assert isinstance(model.country_emb, nn.Embedding)  # small: untouched
    lossf = nn.BCEWithLogitsLoss()
    g = torch.Generator().manual_seed(0)
    for step in range(20):
        user = torch.randint(0, 1_000_000, (256,), generator=g).to(ctx.device)
        item = torch.randint(0, 500_000, (256,), generator=g).to(ctx.device)
        country = torch.randint(0, 32, (256,), generator=g).to(ctx.device)
        labels = (torch.rand(256, generator=g) < 0.3).float().to(ctx.device)
        opt.zero_grad()
        loss = lossf(model(user, item, country), labels)
        loss.backward()
        opt.step()
        if ctx.rank == 0 and (step + 1) % 5 == 0:
            print(f"step {step + 1}: loss={loss.item():.4f}")
    if ctx.rank == 0:
        rows = model.user_emb.variable.sharded.shard.num_rows
        print(f"user_emb rows materialized on rank 0: {rows}")


if __name__ == "__main__":
    main()
