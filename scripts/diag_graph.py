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


emb = model.embedding
keys = sparse + emb.field_offsets
svar = emb.variable.sharded


def pull_only():
    out, h = svar.pull(keys)
    return out.sum()


def pull_push_update():
    out, h = svar.pull(keys)
    svar.push(h, torch.ones_like(out))
    svar.update_weights()
    return out.sum()


static_emb = torch.randn(4096, 26, 10, device=DEV)


def mlp_only():
    opt.zero_grad(set_to_none=False)
    e = static_emb[..., :9]
    lin = static_emb[..., 9]
    s = e.sum(dim=1)
    fm2 = 0.5 * (s * s - (e * e).sum(dim=1)).sum(dim=1)
    deep_in = torch.cat([e.flatten(1), dense], dim=1)
    with torch.autocast("cuda", dtype=torch.bfloat16, enabled=AMP,
                        cache_enabled=False):
        out = model.dnn(deep_in).squeeze(-1)
    logit = (lin.sum(dim=1) + model.dense_linear(dense).squeeze(-1)
             + fm2 + out.float())
    loss = lossf(logit, labels)
    loss.backward()
    opt.optimizer.step()
    return loss


stage = sys.argv[1] if len(sys.argv) > 1 else "full"
fn = {"full": full_step, "fwd": fwd_only, "fwdbwd": fwd_bwd,
      "pull": pull_only, "pullpush": pull_push_update,
      "mlp": mlp_only}[stage.replace("fresh", "")]
FRESH = "fresh" in stage
if stage.startswith(("pull", "pullpush")):
    svar.set_optimizer("adagrad", learning_rate=0.005)

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


def refresh(i):
    """Copy a fresh batch into the static tensors (what bench.py does)."""
    if not FRESH:
        return
    d2, s2, l2 = synthetic_batch(4096, generator=torch.Generator().manual_seed(100 + i))
    dense.copy_(d2.to(DEV))
    sparse.copy_(s2.to(DEV))
    labels.copy_(l2.to(DEV))
    log(f"  fresh batch {i} copied")


for i in range(5):
    refresh(i)
    graph.replay()
    torch.cuda.synchronize()
    log(f"  replay {i} ok")
log("replay ok")
import time
t0 = time.perf_counter()
for i in range(100):
    refresh(100 + i)
    graph.replay()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 100
log(f"replay {dt*1000:.3f} ms/step -> {4096/dt:.0f} samples/s")
