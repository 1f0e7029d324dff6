#!/usr/bin/env python3
"""2-process RCCL validation on ONE GPU.

Exercises the real nccl(=RCCL) backend end-to-end — all_to_all_single over
the padded route, allreduce of dense grads — with both ranks pinned to
cuda:0 (no 2-GPU box needed). Validates the multi-rank engine state against
a world-1 reference over the union of both ranks' batches.

    python scripts/rccl_2rank_1gpu.py            # launcher, spawns 2 ranks
    (writes gpurun_out/rccl_2rank.log on success/failure)

If RCCL refuses two ranks on one device ("Duplicate GPU detected"), the
failure mode is recorded — that is itself the answer hardware gives.
"""

import json
import os
import socket
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

DIM = 9
VOCAB = 100_000
STEPS = 4
N = 4096


def _batches(rank):
    import torch
    g = torch.Generator().manual_seed(1000 + rank)
    out = []
    for _ in range(STEPS):
        keys = torch.randint(0, VOCAB, (N,), dtype=torch.int64, generator=g)
        grads = torch.randn(N, DIM, generator=g)
        out.append((keys, grads))
    return out


def worker():
    import torch
    import torch.distributed as dist
    from openembedding_amd.context import Context

    rank = int(os.environ["RANK"])
    dist.init_process_group("nccl")
    torch.cuda.set_device(0)
    ctx = Context(device="cuda:0")
    st = ctx.create_storage()
    var = st.create_variable(VOCAB, DIM)
    var.set_initializer("uniform", minval=-1.0, maxval=1.0)
    var.set_optimizer("adagrad", learning_rate=0.05,
                      initial_accumulator_value=0.1, epsilon=1e-10)

    t0 = time.time()
    for keys, grads in _batches(rank):
        out, h = var.pull(keys.cuda())
        var.push(h, grads.cuda())
        st.update_weights()
    torch.cuda.synchronize()
    dist.barrier()
    elapsed = time.time() - t0
    var.check_padded_overflow()

    # dense-allreduce sanity on the real backend
    t = torch.full((1024,), float(rank + 1), device="cuda:0")
    dist.all_reduce(t)
    assert torch.all(t == 3.0), "allreduce mismatch"

    # validate vs a world-1 reference over the union of batches
    probe = torch.unique(torch.cat(
        [k for r in range(2) for k, _ in _batches(r)])).cuda()
    after, _ = var.pull(probe, readonly=True)
    if rank == 0:
        dist.destroy_process_group()
        os.environ["WORLD_SIZE"] = "1"
        ref_ctx = Context(device="cuda:0")
        ref_st = ref_ctx.create_storage()
        ref = ref_st.create_variable(VOCAB, DIM)
        ref.set_initializer("uniform", minval=-1.0, maxval=1.0)
        ref.set_optimizer("adagrad", learning_rate=0.05,
                          initial_accumulator_value=0.1, epsilon=1e-10)
        # interleave in step order: both ranks' batch t pulls, then commits
        batches = [_batches(0), _batches(1)]
        for t_i in range(STEPS):
            hs = []
            for r in range(2):
                keys, grads = batches[r][t_i]
                _, h = ref.pull(keys.cuda())
                hs.append((h, grads))
            for h, grads in hs:
                ref.push(h, grads.cuda())
            ref_st.update_weights()
        ref_after, _ = ref.pull(probe, readonly=True)
        torch.testing.assert_close(after, ref_after, rtol=1e-4, atol=1e-5)
        backend_info = {
            "ok": True,
            "padded_mode": var._use_padded(),
            "steps": STEPS,
            "elapsed_s": elapsed,
            "probe_keys": int(probe.numel()),
            "nccl_version": list(torch.cuda.nccl.version()),
        }
        print("RCCL_2RANK_OK " + json.dumps(backend_info), flush=True)
    else:
        dist.destroy_process_group()


def main():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update(RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        procs.append(subprocess.Popen(
            [sys.executable, os.path.abspath(__file__), "--worker"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    out0, _ = procs[0].communicate(timeout=600)
    out1, _ = procs[1].communicate(timeout=120)
    text = out0.decode() + "\n--- rank1 ---\n" + out1.decode()
    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "rccl_2rank.log"), "w") as f:
        f.write(text)
    print(text[-2000:])
    ok = procs[0].returncode == 0 and procs[1].returncode == 0 \
        and "RCCL_2RANK_OK" in text
    print(f"rccl_2rank_1gpu: {'PASS' if ok else 'FAIL'}")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    if "--worker" in sys.argv:
        worker()
    else:
        main()
