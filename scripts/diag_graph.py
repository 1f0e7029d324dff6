"""Bisect hipGraph capture of the train step (diagnostic, GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import openembedding_amd.torch as embed
from openembedding_amd.models import DeepFM, synthetic_batch

torch.manual_seed(0)
DEV = "cuda:0"


def log(msg):
    print(msg, flush=True)


model = DeepFM(dim=9).to(DEV)
opt = embed.distributed_optimizer(
    torch.optim.Adagrad(model.parameters(), lr=0.005))
lossf = torch.nn.BCEWithLogitsLoss()
g0 = torch.Generator().manual_seed(1)
dense, sparse, labels = synthetic_batch(4096, generator=g0)
dense, sparse, labels = dense.to(DEV), sparse.to(DEV), labels.to(DEV)


AMP = "--amp" in sys.argv


def full_step():
    opt.zero_grad(set_to_none=False)
    with torch.autocast("cuda", dtype=torch.bfloat16, enabled=AMP,
                        cache_enabled=False):
        out = model(dense, sparse)
    loss = lossf(out.float(), labels)
    loss.backward()
    opt.step()
    return loss


def fwd_only():
    return model(dense, sparse).sum()


def fwd_bwd():
    opt.zero_grad(set_to_none=False)
    loss = lossf(model(dense, sparse).float(), labels)
    loss.backward()
    return loss


stage = sys.argv[1] if len(sys.argv) > 1 else "full"
fn = {"full": full_step, "fwd": fwd_only, "fwdbwd": fwd_bwd}[stage]

log(f"stage={stage}: eager warmups")
for _ in range(3):
    fn()
torch.cuda.synchronize()
log("warm ok; side-stream warmup")
side = torch.cuda.Stream()
side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for _ in range(3):
        fn()
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize()
log("side ok; capturing")
graph = torch.cuda.CUDAGraph()
with torch.cuda.graph(graph):
    fn()
log("captured; replaying x5")
for _ in range(5):
    graph.replay()
torch.cuda.synchronize()
log("replay ok")
import time
t0 = time.perf_counter()
for _ in range(100):
    graph.replay()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 100
log(f"replay {dt*1000:.3f} ms/step -> {4096/dt:.0f} samples/s")
