#!/usr/bin/env python3
"""Logistic regression with hash-mode embeddings over the full int64 key
space — the reference's examples/criteo_lr_subclass.py (input_dim=-1 ->
2^63 hash table, num_shards) rebuilt as a torch module.

Feature ids can be ANY int64 (e.g. hashed strings); rows are created lazily
on first touch in the GPU hash table.
"""

import argparse

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))))  # run from a source checkout

import openembedding_amd.torch as embed
from openembedding_amd.models import N_DENSE, synthetic_batch


class HashLR(torch.nn.Module):
    def __init__(self):
        super().__init__()
        # num_embeddings=-1 -> hash mode over the full key space
        self.embedding = embed.Embedding(-1, 1)
        self.dense_linear = torch.nn.Linear(N_DENSE, 1)

    def forward(self, dense, sparse):
        # spread per-field ids across the int64 space like hashed features
        keys = sparse * 0x9E3779B97F4A7C15 + torch.arange(
            sparse.shape[1], device=sparse.device)
        w = self.embedding(keys).squeeze(-1)          # [B, F]
        return w.sum(dim=1) + self.dense_linear(dense).squeeze(-1)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=4096)
    p.add_argument("--steps", type=int, default=50)
    args = p.parse_args()

    ctx = embed.get_context()
    model = HashLR().to(ctx.device)
    # dense params train with Adagrad; the sparse side runs the server FTRL
    # (torch has no FTRL — the engine's sparse optimizer set is the
    # reference's, core/optimizers.py)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01),
        sparse_config=dict(category="ftrl", learning_rate=0.05,
                           l1_regularization_strength=0.001,
                           l2_regularization_strength=0.001))
    lossf = torch.nn.BCEWithLogitsLoss()
    for step in range(args.steps):
        dense, sparse, labels = synthetic_batch(args.batch)
        dense, sparse, labels = (dense.to(ctx.device), sparse.to(ctx.device),
                                 labels.to(ctx.device))
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        if ctx.rank == 0 and (step + 1) % 10 == 0:
            print(f"step {step + 1}: loss={loss.item():.4f} "
                  f"rows={model.embedding.variable.sharded.shard.num_rows}")


if __name__ == "__main__":
    main()
