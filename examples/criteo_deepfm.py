#!/usr/bin/env python3
"""Train DeepFM on Criteo-shaped synthetic data — the 3-line-change demo.

The MI355X analogue of the reference's examples/criteo_deepctr_network.py:
the embedding layers are PS-backed (sharded across GPUs when launched with
torchrun), the dense MLP is data-parallel under RCCL allreduce.

Single GPU / CPU:
    python examples/criteo_deepfm.py --steps 100

All 8 GPUs of one node:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/criteo_deepfm.py --steps 100
"""

import argparse
import time

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))))  # run from a source checkout

import openembedding_amd.torch as embed
from openembedding_amd.models import MODELS, synthetic_batch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="deepfm",
                   choices=["deepfm", "wdl", "xdeepfm", "lr"])
    p.add_argument("--dim", type=int, default=9)
    p.add_argument("--batch", type=int, default=4096)
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--lr", type=float, default=0.005)
    p.add_argument("--checkpoint", default="",
                   help="save server+dense model here at the end")
    p.add_argument("--eval-batches", type=int, default=0,
                   help="after training, report AUC/logloss over this many "
                        "held-out synthetic batches")
    p.add_argument("--data", default="",
                   help="train from a Criteo-format file instead of "
                        "synthetic batches (openembedding_amd.data); "
                        "*.tfrecord selects the TFRecord reader")
    args = p.parse_args()

    ctx = embed.get_context()
    torch.manual_seed(1234)
    kw = {} if args.model == "lr" else {"dim": args.dim}
    model = MODELS[args.model](**kw).to(ctx.device)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=args.lr))
    lossf = torch.nn.BCEWithLogitsLoss()

    gen = torch.Generator().manual_seed(1 + ctx.rank)

    def batches():
        if args.data:
            from openembedding_amd.data import (BackgroundLoader,
                                                CriteoTFRecord, CriteoTSV)
            reader_cls = (CriteoTFRecord
                          if args.data.endswith(".tfrecord")
                          else CriteoTSV)
            # every rank reads the file; rank r trains rows r, r+W, ...
            # (batch-level round-robin — the reference sharded csv the
            # same way in its horovod examples)
            for i, b in enumerate(BackgroundLoader(
                    reader_cls(args.data, args.batch), depth=4)):
                if i % ctx.world_size == ctx.rank:
                    yield b
        else:
            while True:
                yield synthetic_batch(args.batch, generator=gen)

    t0 = time.perf_counter()
    src = batches()
    for step in range(args.steps):
        try:
            dense, sparse, labels = next(src)
        except StopIteration:
            break
        dense, sparse, labels = (dense.to(ctx.device), sparse.to(ctx.device),
                                 labels.to(ctx.device))
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        last = step
        if ctx.rank == 0 and (step + 1) % 20 == 0:
            dt = time.perf_counter() - t0
            sps = args.batch * ctx.world_size * (step + 1) / dt
            print(f"step {step + 1}: loss={loss.item():.4f} "
                  f"{sps:,.0f} samples/s")

    if ctx.rank == 0:
        dt = time.perf_counter() - t0
        print(f"done: {last + 1} steps, final loss={loss.item():.4f}, "
              f"{(last + 1) * args.batch * ctx.world_size / dt:,.0f} samples/s")
    if args.eval_batches:
        # ALL ranks evaluate (model forward is a collective embedding pull
        # at world>1 — a rank-0-only loop would deadlock the others), then
        # sync() merges the per-rank accumulator state.
        from openembedding_amd.models.metrics import StreamingAUC, \
            StreamingLogLoss
        auc, ll = StreamingAUC(), StreamingLogLoss()
        egen = torch.Generator().manual_seed(991 + ctx.rank)
        with torch.no_grad():
            for _ in range(args.eval_batches):
                dense, sparse, labels = synthetic_batch(args.batch,
                                                        generator=egen)
                out = model(dense.to(ctx.device), sparse.to(ctx.device))
                auc.update(out, labels.to(ctx.device))
                ll.update(out, labels.to(ctx.device))
        auc.sync()
        ll.sync()
        if ctx.rank == 0:
            print(f"eval: auc={auc.compute():.4f} "
                  f"logloss={ll.compute():.4f}")

    if args.checkpoint:
        wrapped = embed.Model(model)
        wrapped.save_weights(args.checkpoint)
        if ctx.rank == 0:
            print(f"saved to {args.checkpoint}[.openembedding]")


if __name__ == "__main__":
    main()
